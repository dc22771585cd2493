#!/usr/bin/env python3
"""Microbench: gemm_train (8-phase 256^2 MFMA template) vs hipBLASLt on
the llama2-7b LoRA train-step GEMM shapes (M = mb4 x seq512 = 2048,
NT: C = A @ B^T with B = [N, K] weights).

Rotates fresh weight tensors (L3 is 256 MB) like bench_decode_gemm.
Run on a GPU box: python scripts/bench_train_gemm.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from runbooks_amd import ops

SHAPES = [  # (M, N, K): fwd projections + dgrad-as-NT shapes
    (2048, 12288, 4096),   # fused qkv fwd
    (2048, 4096, 4096),    # o fwd
    (2048, 22016, 4096),   # gate/up fwd
    (2048, 4096, 11008),   # down fwd
    (2048, 11008, 4096),   # down dgrad via W^T copy
    (2048, 32000, 4096),   # lm_head fwd
]


def bench(fn, n=20):
    for _ in range(4):
        fn(0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(i)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    torch.manual_seed(3)
    for M, N, K in SHAPES:
        n_w = max(4, int((1 << 30) // (N * K * 2)) + 1)
        ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev)
              for _ in range(n_w)]
        a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        fl = 2.0 * M * N * K

        t_lib = bench(lambda i: torch.nn.functional.linear(a, ws[i % n_w]))
        row = {"M": M, "N": N, "K": K,
               "hipblaslt_ms": round(t_lib * 1e3, 4),
               "hipblaslt_tf": round(fl / t_lib / 1e12, 1)}
        try:
            t_own = bench(lambda i: ops.ext().train_gemm_nt(a, ws[i % n_w]))
            row["train_gemm_ms"] = round(t_own * 1e3, 4)
            row["train_gemm_tf"] = round(fl / t_own / 1e12, 1)
            row["speedup"] = round(t_lib / t_own, 3)
        except Exception as e:  # shape unsupported
            row["train_gemm"] = f"skipped: {e}"[:80]
        print(json.dumps(row), flush=True)
        del ws
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
