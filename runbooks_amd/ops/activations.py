"""Fused SwiGLU and cross-entropy (gfx950 kernels / torch CPU reference)."""
from __future__ import annotations

import torch

from . import _backend


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, u):
        ctx.save_for_backward(g, u)
        return _backend.ext().swiglu_fwd(g, u)

    @staticmethod
    def backward(ctx, dy):
        g, u = ctx.saved_tensors
        dg, du = _backend.ext().swiglu_bwd(dy.contiguous(), g, u)
        return dg, du


def swiglu(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    """silu(g) * u, fused on GPU."""
    if _backend.use_hip(g):
        return _SwiGLUFn.apply(g.contiguous(), u.contiguous())
    return torch.nn.functional.silu(g) * u


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        loss, lse = _backend.ext().cross_entropy_fwd(logits, targets, ignore_index)
        valid = (targets != ignore_index)
        n_valid = int(valid.sum())
        ctx.save_for_backward(logits, targets, lse)
        ctx.n_valid = max(n_valid, 1)
        ctx.ignore_index = ignore_index
        return loss.sum() / max(n_valid, 1)

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        gscale = float(dloss) / ctx.n_valid
        dlogits = _backend.ext().cross_entropy_bwd(
            logits, targets, lse, gscale, ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                  ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over non-ignored targets. logits [..., V] bf16/f32 (kept in
    native dtype on GPU — no fp32 logits materialization); targets [...]
    int. Returns scalar fp32 loss."""
    flat_l = logits.reshape(-1, logits.shape[-1])
    flat_t = targets.reshape(-1)
    if _backend.use_hip(logits):
        return _CrossEntropyFn.apply(flat_l.contiguous(),
                                     flat_t.to(torch.int32).contiguous(),
                                     ignore_index)
    return torch.nn.functional.cross_entropy(
        flat_l.float(), flat_t.long(), ignore_index=ignore_index)
