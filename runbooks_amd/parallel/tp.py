"""Tensor parallelism over xGMI (RCCL all-reduce / all-gather).

Megatron-style column/row sharding sized for the MI355X node: the 8-GPU
node is fully connected (7 p2p links), so per-layer TP all-reduces on
activation slices are the scaling cost; rows/columns are sharded so each
rank's GEMM stays MFMA-shaped and each all-reduce moves B*S*hidden bf16
bytes once per attention block and once per MLP block.

At world_size 1 (or no process group) every layer degrades to a plain
dense linear, so the same model runs single-GPU and CPU tests unchanged.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
from torch import nn

from . import comm
from ..ops.linear import fast_linear


class _AllReduceFn(torch.autograd.Function):
    """Identity fwd + all-reduce bwd (input of a column-parallel layer)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        if comm.is_dist():
            dy = dy.contiguous()
            dist.all_reduce(dy, group=ctx.group)
        return dy, None


class _ReduceFn(torch.autograd.Function):
    """All-reduce fwd + identity bwd (output of a row-parallel layer)."""

    @staticmethod
    def forward(ctx, x, group):
        if comm.is_dist():
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, dy):
        return dy, None


class _GatherFn(torch.autograd.Function):
    """All-gather along the last dim fwd; slice own shard bwd. Used to
    assemble full-vocab logits from a vocab-parallel lm_head."""

    @staticmethod
    def forward(ctx, x, group):
        if not comm.is_dist():
            return x
        ctx.group = group
        ws = dist.get_world_size(group)
        ctx.rank = dist.get_rank(group)
        ctx.shard = x.shape[-1]
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(ws)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, dy):
        if not comm.is_dist():
            return dy, None
        s = ctx.shard
        lo = ctx.rank * s
        return dy[..., lo:lo + s].contiguous(), None


def gather_from_tp(x, group=None):
    return _GatherFn.apply(x, group)


def copy_to_tp(x, group=None):
    return _AllReduceFn.apply(x, group)


def reduce_from_tp(x, group=None):
    return _ReduceFn.apply(x, group)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T sharded over output features; output stays sharded."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 tp_size: int | None = None, dtype=None, gather_input: bool = True):
        super().__init__()
        self.tp = tp_size if tp_size is not None else comm.world_size()
        assert out_features % self.tp == 0, (out_features, self.tp)
        self.in_features = in_features
        self.out_features = out_features
        self.shard = out_features // self.tp
        self.gather_input = gather_input
        self.weight = nn.Parameter(torch.empty(self.shard, in_features, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.shard, dtype=dtype)) if bias else None

    def forward(self, x):
        if self.tp > 1 and self.gather_input:
            x = copy_to_tp(x)
        return fast_linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """Y = X W^T sharded over input features; output is all-reduced."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 tp_size: int | None = None, dtype=None):
        super().__init__()
        self.tp = tp_size if tp_size is not None else comm.world_size()
        assert in_features % self.tp == 0, (in_features, self.tp)
        self.in_features = in_features
        self.out_features = out_features
        self.shard = in_features // self.tp
        self.weight = nn.Parameter(torch.empty(out_features, self.shard, dtype=dtype))
        # bias added once (after reduce), only on rank 0's addition path
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None

    def forward(self, x):
        y = fast_linear(x, self.weight)
        if self.tp > 1:
            y = reduce_from_tp(y)
        if self.bias is not None:
            y = y + self.bias
        return y
