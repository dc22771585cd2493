#!/usr/bin/env python3
"""paged_decode microbench: effective KV-read bandwidth per shape.

Rotates block tables across a multi-GB cache pool so the KV stream
comes from HBM, not the 256 MB L3 (a single small cache re-read every
iteration measures the cache hierarchy, not the kernel).

GPU box: python scripts/bench_paged_decode.py
Prints one JSON line per (shape, len, dtype).
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def bench(B, Hq, Hkv, Dh, length, fp8, iters=50):
    from runbooks_amd.ops.attention import paged_decode

    BS = 16
    blocks_per_seq = (length + BS - 1) // BS
    # pool sized >= 4 GB so rotated tables never re-read L3-resident rows
    row_bytes = (Dh + 16) if fp8 else Dh * 2
    seq_bytes = 2 * blocks_per_seq * BS * Hkv * row_bytes
    rotate = max(2, (4 << 30) // max(1, seq_bytes * B))
    rotate = min(rotate, 8)
    num_blocks = B * blocks_per_seq * rotate + 1

    if fp8:
        # random e4m3 bytes are fine for a bandwidth measurement; zero
        # the scale+pad tail so no NaN/inf scales enter the softmax
        kc = torch.randint(0, 255, (num_blocks, Hkv, BS, Dh + 16),
                           device="cuda", dtype=torch.uint8)
        kc[..., Dh:] = 0
        kv = kc.clone()
    else:
        kc = torch.randn(num_blocks, Hkv, BS, Dh, device="cuda",
                         dtype=torch.bfloat16)
        kv = torch.randn_like(kc)
        if Hq // Hkv >= 4 and Dh <= 128:  # MFMA path: transposed-V layout
            kv = kv.permute(0, 1, 3, 2).contiguous()

    q = torch.randn(B, Hq, Dh, device="cuda", dtype=torch.bfloat16)
    seq_lens = torch.full((B,), length, device="cuda", dtype=torch.int32)
    tables = []
    for r in range(rotate):
        base = 1 + r * B * blocks_per_seq
        t = torch.arange(base, base + B * blocks_per_seq, device="cuda",
                         dtype=torch.int32).view(B, blocks_per_seq)
        tables.append(t)

    for t in tables:
        paged_decode(q, kc, kv, t, seq_lens)
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ev0.record()
    for i in range(iters):
        paged_decode(q, kc, kv, tables[i % rotate], seq_lens)
    ev1.record()
    torch.cuda.synchronize()
    us = ev0.elapsed_time(ev1) * 1000.0 / iters
    bytes_read = 2 * B * length * Hkv * row_bytes
    return us, bytes_read / (us * 1e-6) / 1e12


def main():
    shapes = [
        ("llama2-7b", 32, 32, 32, 128),
        ("llama2-70b", 32, 64, 8, 128),
        ("falcon-40b-mqa", 32, 128, 8, 64),
        ("gemma-dh256", 32, 16, 16, 256),
    ]
    for name, B, Hq, Hkv, Dh in shapes:
        for length in (128, 512, 2048):
            for fp8 in ((False, True) if name == "llama2-7b" else (False,)):
                us, tbs = bench(B, Hq, Hkv, Dh, length, fp8)
                print(json.dumps({
                    "shape": name, "B": B, "Hq": Hq, "Hkv": Hkv, "Dh": Dh,
                    "len": length, "dtype": "fp8" if fp8 else "bf16",
                    "us": round(us, 2), "kv_tb_s": round(tbs, 2)}),
                    flush=True)


if __name__ == "__main__":
    main()
