#!/usr/bin/env python3
"""Single-shape paged_decode runner for rocprofv3 --pmc counter runs.

GPU box:
  rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      SQ_ACTIVE_INST_VALU SQ_ACTIVE_INST_LDS SQ_LDS_BANK_CONFLICT \
      -d out -- python scripts/pmc_decode.py falcon 2048 2
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

SHAPES = {
    "llama7b": (32, 32, 32, 128),
    "llama70b": (32, 64, 8, 128),
    "falcon": (32, 128, 8, 64),
}


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "falcon"
    length = int(sys.argv[2]) if len(sys.argv) > 2 else 2048
    nsplit = int(sys.argv[3]) if len(sys.argv) > 3 else 2
    B, Hq, Hkv, Dh = SHAPES[name]
    from runbooks_amd.ops.attention import paged_decode
    BS = 16
    bps = (length + BS - 1) // BS
    num_blocks = B * bps * 2 + 1
    kc = torch.randn(num_blocks, Hkv, BS, Dh, device="cuda",
                     dtype=torch.bfloat16)
    kv = torch.randn_like(kc)
    if Hq // Hkv >= 4 and Dh <= 128:   # MFMA path: transposed-V layout
        kv = kv.permute(0, 1, 3, 2).contiguous()
    q = torch.randn(B, Hq, Dh, device="cuda", dtype=torch.bfloat16)
    seq_lens = torch.full((B,), length, device="cuda", dtype=torch.int32)
    t = torch.arange(1, 1 + B * bps, device="cuda",
                     dtype=torch.int32).view(B, bps)
    for _ in range(3):
        paged_decode(q, kc, kv, t, seq_lens, nsplit=nsplit)
    torch.cuda.synchronize()
    for _ in range(10):
        paged_decode(q, kc, kv, t, seq_lens, nsplit=nsplit)
    torch.cuda.synchronize()
    print("done", name, length, nsplit)


if __name__ == "__main__":
    main()
