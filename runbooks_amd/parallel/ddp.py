"""Data parallelism: bucketed gradient all-reduce overlapped with backward.

MI355X-first sizing: xGMI is point-to-point (7 links x ~153 GB/s per
GPU), so a ring all-reduce is bound by ONE link. Buckets default to 64 MiB
— large enough to amortize RCCL launch latency, small enough that several
buckets are in flight while backward is still producing grads, keeping
all 7 links busy via RCCL's fully-connected topology.

Mechanics: a post-accumulate-grad hook moves each finished ``.grad``
into its slot of a pre-allocated flat bucket buffer (autograd would
replace a pre-seeded grad view out-of-place, so the copy is explicit);
the moment a bucket is complete its async all-reduce launches (RCCL's
internal stream) while backward keeps producing the next bucket.
``finish_backward()`` waits, averages, and points every ``.grad`` at its
reduced bucket slice for the optimizer.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
from torch import nn

from . import comm


class _Bucket:
    def __init__(self, params: list[torch.nn.Parameter], dtype, device):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.buffer = torch.zeros(self.numel, dtype=dtype, device=device)
        self.views = []
        off = 0
        for p in params:
            v = self.buffer[off:off + p.numel()].view_as(p)
            self.views.append(v)
            off += p.numel()
        self.pending = 0
        self.work = None


class DataParallel(nn.Module):
    """Wraps a module for DP training. World size 1 -> near-zero overhead."""

    def __init__(self, module: nn.Module, bucket_mb: int = 64):
        super().__init__()
        self.module = module
        self.bucket_bytes = bucket_mb << 20
        self._buckets: list[_Bucket] = []
        self._param_bucket: dict[int, tuple[_Bucket, int]] = {}
        self._hooks = []
        # gradient accumulation: when False, completed buckets do NOT
        # launch their all-reduce (grads keep accumulating locally);
        # the final micro-batch sets it back to True.
        self.sync = True
        self._build_buckets()
        if comm.is_dist():
            self._broadcast_params()

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    # -- setup ---------------------------------------------------------------
    def _build_buckets(self):
        params = [p for p in self.module.parameters() if p.requires_grad]
        # Reverse registration order approximates backward completion order,
        # so early buckets fill (and launch) first during backward.
        params = list(reversed(params))
        cur: list[nn.Parameter] = []
        cur_bytes = 0
        groups: list[list[nn.Parameter]] = []
        for p in params:
            sz = p.numel() * p.element_size()
            if cur and cur_bytes + sz > self.bucket_bytes:
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            groups.append(cur)
        for g in groups:
            b = _Bucket(g, g[0].dtype, g[0].device)
            self._buckets.append(b)
            for i, p in enumerate(g):
                self._param_bucket[id(p)] = (b, i)
                h = p.register_post_accumulate_grad_hook(self._on_grad)
                self._hooks.append(h)
        self._reset_pending()

    def _broadcast_params(self):
        for p in self.module.parameters():
            dist.broadcast(p.data, src=0)

    def _reset_pending(self):
        for b in self._buckets:
            b.pending = len(b.params)
            b.work = None

    # -- backward-side hooks ---------------------------------------------------
    def _on_grad(self, p: torch.nn.Parameter):
        b, i = self._param_bucket[id(p)]
        # accumulate into the flat buffer and release the autograd tensor;
        # micro-batch accumulation keeps adding into the same slot.
        b.views[i].add_(p.grad)
        p.grad = None
        b.pending -= 1
        if b.pending == 0 and comm.is_dist() and self.sync:
            b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM, async_op=True)

    def start_microbatch(self, sync: bool = True):
        """Arm the hooks for one backward pass. sync=False accumulates
        into the flat buckets without reducing (grad-accumulation
        micro-batches); the last micro-batch passes sync=True."""
        self.sync = sync
        self._reset_pending()

    def finish_backward(self):
        """Wait for in-flight reductions, average, and expose ``.grad``.
        Call after backward()."""
        ws = comm.world_size()
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
            if ws > 1:
                b.buffer.div_(ws)
            for p, v in zip(b.params, b.views):
                p.grad = v
        self._reset_pending()

    def grad_buffers(self):
        return [b.buffer for b in self._buckets]

    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002
        for b in self._buckets:
            b.buffer.zero_()
            for p in b.params:
                p.grad = None   # next backward must produce a fresh tensor
        self._reset_pending()
