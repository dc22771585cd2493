"""Distributed process-group helpers (RCCL over xGMI on GPU, gloo on CPU).

One process per GPU; ``torch.distributed`` backend "nccl" IS RCCL on ROCm.
The 8-GPU MI355X node is fully connected point-to-point (7 xGMI links x
~153 GB/s per GPU) — RCCL handles the topology; our job is bucket/slice
sizing (see ddp.py) and keeping collectives off the compute stream.
"""
from __future__ import annotations

import os
from datetime import timedelta

import torch
import torch.distributed as dist


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def rank() -> int:
    return dist.get_rank() if is_dist() else 0


def world_size() -> int:
    return dist.get_world_size() if is_dist() else 1


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", rank()))


def init_from_env(backend: str | None = None, timeout_s: int = 600) -> int:
    """Initialize from torchrun env vars. Returns local rank. No-op if
    WORLD_SIZE is absent/1 and not forced."""
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws <= 1 and not os.environ.get("RANK"):
        return 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, timeout=timedelta(seconds=timeout_s))
    lr = local_rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(lr)
    return lr


def barrier():
    if is_dist():
        dist.barrier()


def all_reduce_sum(t: torch.Tensor, group=None, async_op: bool = False):
    if not is_dist():
        return None
    return dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group, async_op=async_op)


def all_gather_into(out: torch.Tensor, t: torch.Tensor, group=None):
    if not is_dist():
        out.copy_(t)
        return
    dist.all_gather_into_tensor(out, t.contiguous(), group=group)
