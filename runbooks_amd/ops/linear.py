"""Linear dispatch: route decode-shaped matmuls to the skinny GEMM.

Batch-<=32 single-token decode GEMMs are weight-bandwidth bound;
csrc/skinny_gemm.hip streams W at HBM rate where hipBLASLt's tiles
measured 25-45% (profiles/). Everything else (prefill, training) stays
on hipBLASLt via F.linear — the library is the right tool for big GEMMs.
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import _backend

# The custom kernel currently trails hipBLASLt on most decode shapes
# (see profiles/README.md measurements); keep it opt-in until it wins.
_USE_SKINNY = os.environ.get("RB_SKINNY_GEMM", "0") == "1"


def fast_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor | None = None) -> torch.Tensor:
    if (_USE_SKINNY and bias is None and not torch.is_grad_enabled()
            and x.dtype == torch.bfloat16 and _backend.use_hip(x)
            and weight.is_contiguous()):
        k = x.shape[-1]
        m = x.numel() // k
        n = weight.shape[0]
        if m <= 32 and n % 64 == 0 and k % 256 == 0:
            y = _backend.ext().skinny_gemm(x.reshape(m, k).contiguous(),
                                           weight)
            return y.view(*x.shape[:-1], n)
    return F.linear(x, weight, bias)


class Linear(torch.nn.Linear):
    """nn.Linear with the decode-GEMM fast path."""

    def forward(self, x):
        return fast_linear(x, self.weight, self.bias)
