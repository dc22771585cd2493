#!/usr/bin/env python3
"""nsplit sweep for GQA/MQA paged_decode shapes (grid-starved at
B*Hkv=256 WGs). GPU box: python scripts/sweep_decode_nsplit.py"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def bench(B, Hq, Hkv, Dh, length, nsplit, iters=50):
    from runbooks_amd.ops.attention import paged_decode
    BS = 16
    bps = (length + BS - 1) // BS
    row = Dh * 2
    seq_bytes = 2 * bps * BS * Hkv * row
    rotate = min(8, max(2, (4 << 30) // max(1, seq_bytes * B)))
    num_blocks = B * bps * rotate + 1
    kc = torch.randn(num_blocks, Hkv, BS, Dh, device="cuda",
                     dtype=torch.bfloat16)
    kv = torch.randn_like(kc)
    if Hq // Hkv >= 4 and Dh <= 128:   # MFMA path: transposed-V layout
        kv = kv.permute(0, 1, 3, 2).contiguous()
    q = torch.randn(B, Hq, Dh, device="cuda", dtype=torch.bfloat16)
    seq_lens = torch.full((B,), length, device="cuda", dtype=torch.int32)
    tables = [torch.arange(1 + r * B * bps, 1 + (r + 1) * B * bps,
                           device="cuda", dtype=torch.int32).view(B, bps)
              for r in range(rotate)]
    for t in tables:
        paged_decode(q, kc, kv, t, seq_lens, nsplit=nsplit)
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for i in range(iters):
        paged_decode(q, kc, kv, tables[i % rotate], seq_lens, nsplit=nsplit)
    e1.record()
    torch.cuda.synchronize()
    us = e0.elapsed_time(e1) * 1000.0 / iters
    return us, 2 * B * length * Hkv * row / (us * 1e-6) / 1e12


def main():
    shapes = [
        ("llama2-70b", 32, 64, 8, 128),
        ("falcon-40b-mqa", 32, 128, 8, 64),
        ("llama2-70b-b8", 8, 64, 8, 128),
    ]
    for name, B, Hq, Hkv, Dh in shapes:
        for length in (512, 2048):
            row = []
            for ns in (1, 2, 4, 8, 16):
                us, tbs = bench(B, Hq, Hkv, Dh, length, ns)
                row.append((ns, round(us, 1), round(tbs, 2)))
            print(json.dumps({"shape": name, "len": length,
                              "ns_us_tbs": row}), flush=True)


if __name__ == "__main__":
    main()
