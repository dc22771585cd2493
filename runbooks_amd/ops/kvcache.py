"""Paged KV cache tensors + append op (gfx950 kernel / CPU reference)."""
from __future__ import annotations

import torch

from . import _backend

BLOCK_SIZE = 16  # tokens per cache block


def alloc_kv_cache(num_blocks: int, num_kv_heads: int, head_dim: int,
                   device, dtype=torch.bfloat16, block_size: int = BLOCK_SIZE,
                   fp8: bool = False, v_transposed: bool = False):
    """fp8=True: e4m3 cache rows of head_dim bytes + f32 scale (+pad) —
    half the KV bytes of bf16 (decode attention is KV-bandwidth bound at
    long context) and double the capacity within 288 GB.

    v_transposed=True (bf16 only): v blocks are [head_dim, block_size]
    so the MFMA decode kernel reads V^T fragments straight from HBM (the
    LDS-transpose alternative measured 26-43% of wave cycles in bank
    conflicts). kv_append and paged_decode key on the shape."""
    if fp8:
        # dh e4m3 bytes + f32 scale + pad to 16 (16-aligned rows: the
        # decode loop stays on full 16 B/lane loads)
        shape = (num_blocks, num_kv_heads, block_size, head_dim + 16)
        k = torch.zeros(shape, device=device, dtype=torch.uint8)
        if v_transposed:
            # fp8 MFMA path: v blocks [dh+4, bs] bytes — the 4 tail
            # rows are 64 B = the 16 per-token f32 scales
            v = torch.zeros((num_blocks, num_kv_heads, head_dim + 4,
                             block_size), device=device, dtype=torch.uint8)
        else:
            v = torch.zeros(shape, device=device, dtype=torch.uint8)
        return k, v
    shape = (num_blocks, num_kv_heads, block_size, head_dim)
    k = torch.zeros(shape, device=device, dtype=dtype)
    vshape = ((num_blocks, num_kv_heads, head_dim, block_size)
              if v_transposed else shape)
    v = torch.zeros(vshape, device=device, dtype=dtype)
    return k, v


def fp8_quant_row_ref(x: torch.Tensor):
    """Reference per-row e4m3 quantization (rows = last dim): returns
    (bytes uint8 [..., dh], scale f32 [...])."""
    amax = x.float().abs().amax(dim=-1).clamp(min=1e-8)
    scale = amax / 448.0
    q = (x.float() / scale[..., None]).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale


def fp8_dequant_cache_ref(cache: torch.Tensor) -> torch.Tensor:
    """[blocks, hkv, bs, dh+16] uint8 -> [blocks, hkv, bs, dh] f32;
    transposed-V fp8 blocks ([dh+4, bs] bytes) -> [blocks, hkv, dh, bs]
    f32 (the bf16-vt shape, so paged_decode_ref's vt branch applies)."""
    bs = 16
    if cache.shape[-1] == bs and cache.shape[-2] > bs + 16:
        dh = cache.shape[-2] - 4
        data = cache[..., :dh, :].contiguous().view(
            torch.float8_e4m3fn).float()
        scales = cache[..., dh:, :].reshape(*cache.shape[:-2], -1)
        scales = scales.contiguous().view(torch.float32)  # [..., bs]
        return data * scales[..., None, :]
    dh = cache.shape[-1] - 16
    data = cache[..., :dh].contiguous().view(torch.float8_e4m3fn).float()
    scale = cache[..., dh:dh + 4].contiguous().view(torch.float32)
    return data * scale


def _kv_append_fp8_ref(k, v, k_cache, v_cache, slot_mapping):
    bs = k_cache.shape[2]
    dh = k_cache.shape[-1] - 16
    vt = v_cache.shape[2] == dh + 4 and v_cache.shape[3] == bs
    for t in range(k.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        blk, off = divmod(slot, bs)
        q, scale = fp8_quant_row_ref(k[t])            # [hkv, dh], [hkv]
        k_cache[blk, :, off, :dh] = q
        k_cache[blk, :, off, dh:dh + 4] = scale.float().view(
            torch.uint8).reshape(-1, 4)
        q, scale = fp8_quant_row_ref(v[t])
        if vt:
            v_cache[blk, :, :dh, off] = q
            # scale of token `off` lives at flat bytes dh*bs + off*4
            tail = v_cache[blk, :, dh:, :].reshape(v_cache.shape[1], -1)
            tail[:, off * 4:off * 4 + 4] = scale.float().view(
                torch.uint8).reshape(-1, 4)
        else:
            v_cache[blk, :, off, :dh] = q
            v_cache[blk, :, off, dh:dh + 4] = scale.float().view(
                torch.uint8).reshape(-1, 4)


def kv_append_ref(k, v, k_cache, v_cache, slot_mapping):
    if k_cache.dtype == torch.uint8:
        return _kv_append_fp8_ref(k, v, k_cache, v_cache, slot_mapping)
    bs = k_cache.shape[2]
    vt = v_cache.shape[2] == k_cache.shape[3] and \
        v_cache.shape[3] == bs and k_cache.shape[3] != bs
    for t in range(k.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        blk, off = divmod(slot, bs)
        k_cache[blk, :, off] = k[t]
        if vt:
            v_cache[blk, :, :, off] = v[t]
        else:
            v_cache[blk, :, off] = v[t]


def kv_append(k: torch.Tensor, v: torch.Tensor, k_cache: torch.Tensor,
              v_cache: torch.Tensor, slot_mapping: torch.Tensor) -> None:
    """k, v: [tokens, Hkv, Dh]; slot_mapping: [tokens] int32 (slot = blk*BS+off)."""
    if _backend.use_hip(k):
        _backend.ext().kv_append(
            k.contiguous(), v.contiguous(), k_cache, v_cache,
            slot_mapping.to(torch.int32),
        )
    else:
        kv_append_ref(k, v, k_cache, v_cache, slot_mapping)
