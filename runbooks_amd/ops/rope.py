"""Rotary position embeddings (rotate-half / neox convention).

cos/sin tables are precomputed on the host (fp32, [max_pos, Dh/2]) — the
gfx950 kernel stays memory-bound instead of burning VALU on sin/cos
(guide: Appendix B, element-wise trig ops).
"""
from __future__ import annotations

import torch

from . import _backend


def rope_tables(
    head_dim: int,
    max_pos: int,
    theta: float = 10000.0,
    device=None,
    scaling: float = 1.0,
) -> tuple[torch.Tensor, torch.Tensor]:
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64) / scaling
    ang = torch.outer(pos, inv_freq)
    cos = ang.cos().float()
    sin = ang.sin().float()
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


def rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             positions: torch.Tensor) -> torch.Tensor:
    """Reference: x [..., T, H, Dh] or [T, H, Dh]; positions [T] int."""
    dh = x.shape[-1]
    half = dh // 2
    c = cos[positions.long()].to(torch.float32)  # [T, half]
    s = sin[positions.long()].to(torch.float32)
    shape = [1] * (x.dim() - 3) + [x.shape[-3], 1, half]
    c = c.reshape(shape)
    s = s.reshape(shape)
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return out.to(x.dtype)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, positions, heads):
        ctx.save_for_backward(cos, sin, positions)
        ctx.heads = heads
        return _backend.ext().rope_fwd(x, cos, sin, positions, heads)

    @staticmethod
    def backward(ctx, dy):
        cos, sin, positions = ctx.saved_tensors
        dx = _backend.ext().rope_bwd(dy.contiguous(), cos, sin, positions, ctx.heads)
        return dx, None, None, None, None


def rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
         positions: torch.Tensor) -> torch.Tensor:
    """Apply RoPE to x of shape [T, H, Dh] (token-major) with positions [T]."""
    assert x.dim() == 3, "rope expects [tokens, heads, head_dim]"
    if _backend.use_hip(x):
        return _RopeFn.apply(
            x.contiguous(), cos, sin, positions.to(torch.int32), x.shape[1]
        )
    return rope_ref(x, cos, sin, positions)
