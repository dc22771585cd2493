"""Attention ops.

* ``causal_attention``: batch-uniform causal GQA attention used by the
  trainer forward/backward. Built from explicit GEMMs (hipBLASLt/rocBLAS
  via torch.matmul — plain library GEMMs) + fused softmax, chunked over
  query blocks to bound the S^2 score memory. No Triton, no SDPA dispatch.
  A hand-written MFMA flash-attention prefill kernel supersedes this on
  the serving path (see csrc/attention_prefill.hip).

* ``paged_decode``: single-token decode against the paged KV cache —
  gfx950 HIP kernel on GPU (csrc/attention_decode.hip), fp32 reference on
  CPU (also the numerics oracle for the GPU test).
"""
from __future__ import annotations

import math

import torch

from . import _backend


class _FlashAttnTrain(torch.autograd.Function):
    """Differentiable causal GQA attention on the gfx950 MFMA kernels.

    Forward: flash_fwd_train (attention_prefill.hip, saves LSE).
    Backward: fa_bwd (attention_bwd.hip, FA2-style recompute); GQA dk/dv
    come back per-q-head and are group-summed to the kv heads here.
    """

    @staticmethod
    def forward(ctx, q, k, v, scale):
        qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
        out, lse = _backend.ext().flash_fwd_train(qc, kc, vc, float(scale))
        ctx.save_for_backward(qc, kc, vc, out, lse)
        ctx.scale = float(scale)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = _backend.ext().fa_bwd(dout.contiguous(), q, k, v, out,
                                           lse, ctx.scale)
        hq, hkv = q.shape[2], k.shape[2]
        if hkv != hq:  # sum the q-head group (GQA/MQA)
            g = hq // hkv
            b, s = dk.shape[0], dk.shape[1]
            dk = dk.view(b, s, hkv, g, -1).sum(dim=3)
            dv = dv.view(b, s, hkv, g, -1).sum(dim=3)
        return dq, dk, dv, None


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: float | None = None, q_block: int = 256) -> torch.Tensor:
    """q [B, S, Hq, Dh]; k, v [B, S, Hkv, Dh] -> [B, S, Hq, Dh].

    Differentiable. GPU bf16 (Dh 64/128) runs the MFMA flash kernels;
    otherwise the chunked fp32 reference below (also the numerics oracle
    for the GPU test). Dh=256 has an MFMA INFERENCE prefill
    (flash_prefill) but trains on the reference (fa_bwd has no 256
    tile yet).
    """
    if _backend.use_hip(q) and q.shape[-1] in (64, 128) \
            and q.dtype == torch.bfloat16:
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        return _FlashAttnTrain.apply(q, k, v, scale)
    return causal_attention_ref(q, k, v, scale=scale, q_block=q_block)


def causal_attention_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                         scale: float | None = None,
                         q_block: int = 256) -> torch.Tensor:
    """Chunked eager reference (fp32 softmax), differentiable."""
    B, S, Hq, Dh = q.shape
    Hkv = k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(Dh)
    g = Hq // Hkv
    if g > 1:
        k = k.repeat_interleave(g, dim=2)
        v = v.repeat_interleave(g, dim=2)
    qt = q.transpose(1, 2)  # [B, H, S, D]
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)

    outs = []
    for s0 in range(0, S, q_block):
        s1 = min(S, s0 + q_block)
        qb = qt[:, :, s0:s1]                              # [B,H,bs,D]
        scores = torch.matmul(qb, kt[:, :, :s1].transpose(-1, -2)).float() * scale
        mask = torch.ones(s1 - s0, s1, dtype=torch.bool, device=q.device)
        mask = torch.triu(mask, diagonal=s0 + 1)
        scores = scores.masked_fill(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1).to(vt.dtype)
        outs.append(torch.matmul(p, vt[:, :, :s1]))
    return torch.cat(outs, dim=2).transpose(1, 2).contiguous()


def flash_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  scale: float | None = None) -> torch.Tensor:
    """Causal GQA attention, inference-only (no autograd).

    q [B, S, Hq, Dh] bf16 on GPU -> gfx950 MFMA flash kernel
    (csrc/attention_prefill.hip); CPU falls back to the chunked reference.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _backend.use_hip(q) and q.shape[-1] in (64, 128, 256) \
            and q.dtype == torch.bfloat16:
        return _backend.ext().flash_prefill(
            q.contiguous(), k.contiguous(), v.contiguous(), float(scale))
    return causal_attention_ref(q, k, v, scale=scale)


def paged_decode_ref(q, k_cache, v_cache, block_tables, seq_lens, scale,
                     seq_starts=None):
    """fp32 reference decode: q [B, Hq, Dh]. fp8 caches (uint8 rows of
    Dh bytes + f32 scale) are dequantized up front. seq_starts (strict
    sliding window) bounds attention to virtual positions
    [start, seq_len)."""
    if k_cache.dtype == torch.uint8:
        from .kvcache import fp8_dequant_cache_ref
        k_cache = fp8_dequant_cache_ref(k_cache)
        v_cache = fp8_dequant_cache_ref(v_cache)
    B, Hq, Dh = q.shape
    _, Hkv, BS, _ = k_cache.shape
    g = Hq // Hkv
    vt = (v_cache.shape[2] == Dh and v_cache.shape[3] == BS
          and Dh != BS)  # transposed-V layout (alloc_kv_cache v_transposed)
    out = torch.empty_like(q)
    for b in range(B):
        n = int(seq_lens[b])
        st = int(seq_starts[b]) if seq_starts is not None else 0
        blocks = block_tables[b, : (n + BS - 1) // BS].long()
        k = k_cache[blocks].transpose(1, 2).reshape(-1, Hkv, Dh)[st:n].float()
        if vt:
            v = v_cache[blocks].permute(0, 3, 1, 2).reshape(
                -1, Hkv, Dh)[st:n].float()
        else:
            v = v_cache[blocks].transpose(1, 2).reshape(
                -1, Hkv, Dh)[st:n].float()
        for h in range(Hq):
            hk = h // g
            s = (k[:, hk] @ q[b, h].float()) * scale
            p = torch.softmax(s, dim=0)
            out[b, h] = (p @ v[:, hk]).to(q.dtype)
    return out


def _is_vt(k_cache, v_cache) -> bool:
    """Transposed-V cache layout (alloc_kv_cache v_transposed) == the
    MFMA decode path; the shape is the routing signal. fp8 vt blocks
    are [dh+4, bs] bytes (scale tail)."""
    if v_cache.dim() != 4:
        return False
    bs = k_cache.shape[2]
    if k_cache.dtype == torch.uint8:
        dh = k_cache.shape[3] - 16
        return v_cache.shape[2] == dh + 4 and v_cache.shape[3] == bs
    return (v_cache.shape[2] == k_cache.shape[3]
            and v_cache.shape[3] == bs and k_cache.shape[3] != bs)


def _auto_nsplit(B, hkv, seq_lens, mfma=False):
    """Work-split (measured sweeps: profiles/decode_attn_pipeline.md).
    The MFMA path is block-granular and latency-light: ~512 WGs is the
    knee. The scalar kernels want ~1024 (4 WGs/CU). Chunks should still
    cover >= ~64 tokens each."""
    base = max(1, B * hkv)
    target = 512 if mfma else 1024
    if base >= target:
        return 1
    max_len = int(seq_lens.max())
    return max(1, min(16, target // base, (max_len + 63) // 64))


def paged_decode(q, k_cache, v_cache, block_tables, seq_lens,
                 scale: float | None = None, nsplit: int | None = None,
                 seq_starts=None):
    """Decode attention over the paged cache. q [B, Hq, Dh] bf16.
    seq_starts (int32 [B], optional): strict-window start positions."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _backend.use_hip(q):
        if nsplit is None:
            nsplit = _auto_nsplit(q.shape[0], k_cache.shape[1], seq_lens,
                                  mfma=_is_vt(k_cache, v_cache))
        return _backend.ext().paged_decode(
            q.contiguous(), k_cache, v_cache, block_tables, seq_lens,
            int(nsplit), float(scale), seq_starts=seq_starts,
        )
    return paged_decode_ref(q, k_cache, v_cache, block_tables, seq_lens,
                            scale, seq_starts=seq_starts)


def paged_decode_with_operand(q, k_cache, v_cache, block_tables, seq_lens,
                              scale: float | None = None,
                              nsplit: int | None = None, seq_starts=None):
    """paged_decode that ALSO returns the decode-GEMM operand layout of
    the attention output (the o_proj input swizzle fused into the
    epilogue) when the kernel emits it; (out, swz_or_None)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _backend.use_hip(q):
        if nsplit is None:
            nsplit = _auto_nsplit(q.shape[0], k_cache.shape[1], seq_lens,
                                  mfma=_is_vt(k_cache, v_cache))
        res = _backend.ext().paged_decode_swz(
            q.contiguous(), k_cache, v_cache, block_tables, seq_lens,
            int(nsplit), float(scale), seq_starts=seq_starts)
        return res[0], (res[1] if len(res) > 1 else None)
    return paged_decode_ref(q, k_cache, v_cache, block_tables, seq_lens,
                            scale, seq_starts=seq_starts), None
