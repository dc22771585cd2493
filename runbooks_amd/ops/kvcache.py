"""Paged KV cache tensors + append op (gfx950 kernel / CPU reference)."""
from __future__ import annotations

import torch

from . import _backend

BLOCK_SIZE = 16  # tokens per cache block


def alloc_kv_cache(num_blocks: int, num_kv_heads: int, head_dim: int,
                   device, dtype=torch.bfloat16, block_size: int = BLOCK_SIZE):
    shape = (num_blocks, num_kv_heads, block_size, head_dim)
    k = torch.zeros(shape, device=device, dtype=dtype)
    v = torch.zeros(shape, device=device, dtype=dtype)
    return k, v


def kv_append_ref(k, v, k_cache, v_cache, slot_mapping):
    bs = k_cache.shape[2]
    for t in range(k.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        blk, off = divmod(slot, bs)
        k_cache[blk, :, off] = k[t]
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
