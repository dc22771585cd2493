"""Token sampling: greedy argmax / Gumbel-max temperature sampling."""
from __future__ import annotations

import torch

from . import _backend


def sample_tokens(logits: torch.Tensor, temperature: float = 0.0,
                  seed: int = 0, top_p: float = 1.0) -> torch.Tensor:
    """logits [..., V] -> int32 token ids [...]. temperature<=0 => greedy;
    0<top_p<1 applies nucleus filtering before sampling."""
    shape = logits.shape[:-1]
    flat = logits.reshape(-1, logits.shape[-1])
    if temperature > 0.0 and 0.0 < top_p < 1.0:
        flat = top_p_filter(flat.float(), top_p)
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


def top_p_filter(logits: torch.Tensor, top_p: float) -> torch.Tensor:
    """Nucleus filtering: keep the smallest prefix of the sorted
    distribution with cumulative probability >= top_p; the rest -> -inf.
    Runs in torch (sampling happens OUTSIDE the decode hipGraph, so the
    sort here never lands in the captured stream)."""
    sorted_logits, idx = torch.sort(logits, dim=-1, descending=True)
    probs = torch.softmax(sorted_logits, dim=-1)
    cum = probs.cumsum(dim=-1)
    # shift so the token that crosses top_p stays included
    drop = cum - probs > top_p
    masked = sorted_logits.masked_fill(drop, float("-inf"))
    out = torch.full_like(logits, float("-inf"))
    return out.scatter(-1, idx, masked)
