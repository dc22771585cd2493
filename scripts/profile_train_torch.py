#!/usr/bin/env python3
"""torch.profiler attribution of one llama2-7b LoRA train step: which
host-side ops own the device time (incl. rocclr copyBuffer / elementwise
adds that a kernel-level rocprof trace can't attribute).

GPU box: python scripts/profile_train_torch.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from runbooks_amd.train import TrainConfig, Trainer
from runbooks_amd.train.data import SyntheticTokens


def main():
    assert torch.cuda.is_available()
    cfg = TrainConfig(model="llama2-7b", seq_len=512, micro_batch=4,
                      num_train_steps=8, dtype="bfloat16", save_steps=0,
                      seed=17)
    tr = Trainer(cfg)
    ds = SyntheticTokens(tr.model.cfg.vocab_size, 513, n=64, seed=0)
    batch = torch.stack([ds[i] for i in range(4)])
    for _ in range(3):
        tr.train_step(batch)
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        for _ in range(2):
            tr.train_step(batch)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="self_cuda_time_total", row_limit=40, max_src_column_width=60))


if __name__ == "__main__":
    main()
