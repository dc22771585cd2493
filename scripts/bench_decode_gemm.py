#!/usr/bin/env python3
"""Microbench: decode_gemm (v2 weight-stream kernel) vs hipBLASLt
(F.linear) on the llama2-7b decode shapes.

Methodology (NOTES-ROUND2 gotcha: isolated GEMM microbenches lie — a
reused weight sits in the 256 MB L3): rotate through >= 8 fresh weight
tensors per shape so every timed call streams from HBM, like the real
decode loop where 13+ GB of weights pass between reuses.

Usage (GPU box): python scripts/bench_decode_gemm.py [--m 32]
Prints one line per shape: ms + effective TB/s for both paths.
"""
import argparse
import json
import time

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from runbooks_amd import ops

SHAPES = [  # (N, K) llama2-7b decode weights
    (12288, 4096),   # fused qkv
    (4096, 4096),    # o_proj
    (22016, 4096),   # fused gate/up
    (4096, 11008),   # down_proj
    (32000, 4096),   # lm_head
]


def bench(fn, n_iter=30):
    for _ in range(5):
        fn(0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n_iter):
        fn(i)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n_iter


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=32)
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()
    assert torch.cuda.is_available()
    dev = "cuda:0"
    torch.manual_seed(7)
    results = []
    for N, K in SHAPES:
        n_w = max(8, int((1 << 30) // (N * K * 2)) + 1)  # >1 GB rotation
        ws = [torch.randn(N, K, dtype=torch.bfloat16, device=dev)
              for _ in range(n_w)]
        x = torch.randn(args.m, K, dtype=torch.bfloat16, device=dev)

        t_lib = bench(lambda i: F.linear(x, ws[i % n_w]), args.iters)
        swz = [ops.ext().decode_swizzle_w(w) for w in ws]
        xs = ops.ext().decode_swizzle_x(x)
        t_v2 = bench(lambda i: ops.ext().decode_gemm(
            xs, swz[i % n_w], args.m, N, K), args.iters)
        t_v2_full = bench(lambda i: ops.ext().decode_gemm(
            ops.ext().decode_swizzle_x(x), swz[i % n_w], args.m, N, K),
            args.iters)
        if os.environ.get("RB_BENCH_SPLIT_SWEEP") == "1":
            for fs in (1, 2, 4):
                if (K // 16) % (fs * 4) or K // (fs * 2) < 1024:
                    continue
                t_fs = bench(lambda i: ops.ext().decode_gemm(
                    xs, swz[i % n_w], args.m, N, K, force_split=fs),
                    args.iters)
                print(json.dumps({
                    "N": N, "K": K, "force_split": fs,
                    "ms": round(t_fs * 1e3, 4),
                    "tbps": round(N * K * 2 / 1e9 / t_fs / 1e3, 3)}),
                    flush=True)
        del swz
        gb = N * K * 2 / 1e9
        row = {
            "N": N, "K": K, "M": args.m, "weights_gb": round(gb, 3),
            "hipblaslt_ms": round(t_lib * 1e3, 4),
            "hipblaslt_tbps": round(gb / t_lib / 1e3, 3),
            "decode_gemm_ms": round(t_v2 * 1e3, 4),
            "decode_gemm_tbps": round(gb / t_v2 / 1e3, 3),
            "with_xswizzle_ms": round(t_v2_full * 1e3, 4),
            "speedup": round(t_lib / t_v2_full, 3),
            "split": int(ops.ext().decode_gemm_split(N, K)),
        }
        results.append(row)
        print(json.dumps(row), flush=True)
        del ws
        torch.cuda.empty_cache()

    tot_lib = sum(r["hipblaslt_ms"] for r in results[:4])
    tot_v2 = sum(r["decode_gemm_ms"] for r in results[:4])
    print(json.dumps({"per_layer_sum_ms": {"hipblaslt": round(tot_lib, 4),
                                           "decode_gemm": round(tot_v2, 4)},
                      "layer_speedup": round(tot_lib / tot_v2, 3)}),
          flush=True)


if __name__ == "__main__":
    main()
