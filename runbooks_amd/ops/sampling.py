"""Token sampling: greedy argmax / Gumbel-max temperature sampling."""
from __future__ import annotations

import torch

from . import _backend


def sample_tokens(logits: torch.Tensor, temperature: float = 0.0,
                  seed: int = 0) -> torch.Tensor:
    """logits [..., V] -> int32 token ids [...]. temperature<=0 => greedy."""
    shape = logits.shape[:-1]
    flat = logits.reshape(-1, logits.shape[-1])
    if _backend.use_hip(logits):
        out = _backend.ext().sample_tokens(flat.contiguous(), float(temperature),
                                           int(seed))
        return out.reshape(shape)
    if temperature <= 0.0:
        return flat.argmax(dim=-1).to(torch.int32).reshape(shape)
    gen = torch.Generator(device="cpu").manual_seed(seed)
    u = torch.rand(flat.shape, generator=gen)
    g = -torch.log(-torch.log(u.clamp_min(1e-20)))
    return (flat.float() / temperature + g).argmax(dim=-1).to(torch.int32).reshape(shape)
