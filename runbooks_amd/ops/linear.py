"""Linear dispatch: route decode-shaped matmuls to the skinny GEMM.

Batch-<=32 single-token decode GEMMs are weight-bandwidth bound;
csrc/skinny_gemm.hip streams W at HBM rate where hipBLASLt's tiles
measured 25-45% (profiles/). Everything else (prefill, training) stays
on hipBLASLt via F.linear — the library is the right tool for big GEMMs.
"""
from __future__ import annotations

import os
import weakref

import torch
import torch.nn.functional as F

from . import _backend

# Round-1 tile kernel; superseded by decode_gemm (kept for A/B runs).
_USE_SKINNY = os.environ.get("RB_SKINNY_GEMM", "0") == "1"
# v2 weight-stream kernel (csrc/decode_gemm.hip): default-on; RB_DECODE_GEMM=0
# falls back to hipBLASLt.
_USE_DECODE_GEMM = os.environ.get("RB_DECODE_GEMM", "1") == "1"


# data_ptr(weight) -> (w8 uint8 view, f32 per-channel scale). Populated by
# quantize_fp8 (serving with MODEL_LOAD_IN_8BIT); looked up on the decode
# fast path.
_FP8_REGISTRY: dict[int, tuple[torch.Tensor, torch.Tensor]] = {}

# data_ptr(weight) -> fragment-lane-major copy ([N/32][K/16][lane][8])
# consumed by csrc/decode_gemm.hip. A second full-precision weight copy:
# 288 GB HBM3E per GPU makes layout-specialized decode weights the right
# trade on MI355X (prefill/training keep the original [N,K] tensor).
# value: (swizzled copy, (N, K), weakref to the registered tensor).
# data_ptr alone is NOT a safe key — a freed weight's address gets
# recycled by later allocations, so every lookup verifies tensor
# identity through the weakref (stale entries are dropped lazily).
_DECODE_W_REGISTRY: dict[int, tuple] = {}


def _registry_get(reg: dict, weight: torch.Tensor):
    entry = reg.get(weight.data_ptr())
    if entry is None:
        return None
    if entry[-1]() is not weight:
        del reg[weight.data_ptr()]
        return None
    return entry


def register_decode_weight(weight: torch.Tensor) -> None:
    """Build + register the decode-GEMM weight layout for `weight`."""
    for ptr in [p for p, e in _DECODE_W_REGISTRY.items() if e[-1]() is None]:
        del _DECODE_W_REGISTRY[ptr]          # purge dead entries
    ws = _backend.ext().decode_swizzle_w(weight.data)
    _DECODE_W_REGISTRY[weight.data_ptr()] = (
        ws, (int(weight.shape[0]), int(weight.shape[1])),
        weakref.ref(weight))

E4M3_MAX = 448.0


@torch.no_grad()
def quantize_fp8(weight: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Row-wise (per-output-channel) OCP e4m3 quantization; registers the
    pair so fast_linear dispatches to the fp8 decode GEMM. Pass the
    PERSISTENT tensor object (Parameter / fused buffer), not .data —
    lookups verify identity via weakref."""
    scale = weight.float().abs().amax(dim=1).clamp(min=1e-8) / E4M3_MAX
    w8 = (weight.float() / scale[:, None]).to(torch.float8_e4m3fn)
    w8_bytes = w8.view(torch.uint8).contiguous()
    _FP8_REGISTRY[weight.data_ptr()] = (w8_bytes, scale.contiguous(),
                                        weakref.ref(weight))
    return w8_bytes, scale


def dequantize_fp8(w8_bytes: torch.Tensor, scale: torch.Tensor,
                   dtype=torch.bfloat16) -> torch.Tensor:
    return (w8_bytes.view(torch.float8_e4m3fn).float() *
            scale[:, None]).to(dtype)


def fast_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor | None = None) -> torch.Tensor:
    if (bias is None and not torch.is_grad_enabled()
            and x.dtype == torch.bfloat16 and _backend.use_hip(x)
            and weight.is_contiguous()):
        k = x.shape[-1]
        m = x.numel() // k
        n = weight.shape[0]
        if m <= 32:
            if n % 64 == 0 and k % 256 == 0:
                q = _registry_get(_FP8_REGISTRY, weight)
                if q is not None:
                    y = _backend.ext().skinny_gemm_fp8(
                        x.reshape(m, k).contiguous(), q[0], q[1])
                    return y.view(*x.shape[:-1], n)
            if _USE_DECODE_GEMM:
                dw = _registry_get(_DECODE_W_REGISTRY, weight)
                if dw is not None:
                    # producers (rmsnorm_fwd_dec, *_packed_dec) attach the
                    # pre-swizzled operand; otherwise one tiny swizzle
                    # kernel builds it here
                    xs = getattr(x, "_rb_swz", None)
                    if xs is None:
                        xs = _backend.ext().decode_swizzle_x(
                            x.reshape(m, k).contiguous())
                    y = _backend.ext().decode_gemm(xs, dw[0], m, n, k)
                    return y.view(*x.shape[:-1], n)
            if _USE_SKINNY and n % 64 == 0 and k % 256 == 0:
                y = _backend.ext().skinny_gemm(x.reshape(m, k).contiguous(),
                                               weight)
                return y.view(*x.shape[:-1], n)
    return F.linear(x, weight, bias)


def decode_linear_raw(x, weight):
    """Decode GEMM that may return an UNCOMBINED fp32 split-K slab
    [split, M, N] (consumer kernels fold it — rmsnorm_res_slab_fwd_dec),
    or a plain bf16 [M, N]. Falls back to fast_linear when the weight
    has no registered decode layout. Returns (out, is_slab)."""
    dw = _registry_get(_DECODE_W_REGISTRY, weight)
    if dw is None or not _USE_DECODE_GEMM:
        with torch.no_grad():
            return fast_linear(x, weight), False
    k = x.shape[-1]
    m = x.numel() // k
    n = weight.shape[0]
    xs = getattr(x, "_rb_swz", None)
    if xs is None:
        xs = _backend.ext().decode_swizzle_x(x.reshape(m, k).contiguous())
    out = _backend.ext().decode_gemm_raw(xs, dw[0], m, n, k)
    return out, out.dtype == torch.float32


class Linear(torch.nn.Linear):
    """nn.Linear with the decode-GEMM fast path."""

    def forward(self, x):
        return fast_linear(x, self.weight, self.bias)
