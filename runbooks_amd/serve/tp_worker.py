"""Tensor-parallel serving: rank-0 drive / worker follow protocol.

Rank 0 runs the scheduler + HTTP server; worker ranks must enter the
same model.prefill/decode calls so the RCCL all-reduces inside the TP
layers (parallel/tp.py) line up. Before each model invocation rank 0
broadcasts a small header (op, batch, seq, table width) and the input
tensors over the same process group; workers replay the call against
their own weight shards and KV caches and discard the logits (the
vocab-parallel lm_head all-gather assembles full logits on every rank,
so the collective schedule stays symmetric).

Ops: 1 = prefill, 2 = decode, 0 = shutdown.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

OP_SHUTDOWN = 0
OP_PREFILL = 1
OP_DECODE = 2


def _bcast(t: torch.Tensor) -> torch.Tensor:
    dist.broadcast(t, src=0)
    return t


def broadcast_prefill(tokens, positions, slots):
    """Called by rank 0's engine before model.prefill."""
    dev = tokens.device
    hdr = torch.tensor([OP_PREFILL, tokens.shape[0], tokens.shape[1], 0],
                       dtype=torch.int64, device=dev)
    _bcast(hdr)
    _bcast(tokens)
    _bcast(positions)
    _bcast(slots)


def broadcast_decode(tokens, positions, slots, block_tables, seq_lens,
                     seq_starts):
    dev = tokens.device
    hdr = torch.tensor([OP_DECODE, tokens.shape[0], 0,
                        block_tables.shape[1]], dtype=torch.int64,
                       device=dev)
    _bcast(hdr)
    _bcast(tokens)
    _bcast(positions)
    _bcast(slots)
    _bcast(block_tables)
    _bcast(seq_lens)
    _bcast(seq_starts)


def broadcast_shutdown(device):
    hdr = torch.tensor([OP_SHUTDOWN, 0, 0, 0], dtype=torch.int64,
                       device=device)
    _bcast(hdr)


def worker_loop(engine) -> None:
    """Worker ranks: follow rank 0's model invocations forever."""
    model = engine.model
    caches = engine.caches
    dev = next(model.parameters()).device
    while True:
        hdr = _bcast(torch.zeros(4, dtype=torch.int64, device=dev))
        op, b, s, maxb = (int(x) for x in hdr)
        if op == OP_SHUTDOWN:
            return
        if op == OP_PREFILL:
            tokens = _bcast(torch.zeros(b, s, dtype=torch.long, device=dev))
            positions = _bcast(torch.zeros(s, dtype=torch.int32, device=dev))
            slots = _bcast(torch.zeros(s, dtype=torch.int32, device=dev))
            model.prefill(tokens, positions, caches, slots)
        elif op == OP_DECODE:
            tokens = _bcast(torch.zeros(b, dtype=torch.long, device=dev))
            positions = _bcast(torch.zeros(b, dtype=torch.int32, device=dev))
            slots = _bcast(torch.zeros(b, dtype=torch.int32, device=dev))
            bt = _bcast(torch.zeros(b, maxb, dtype=torch.int32, device=dev))
            seq_lens = _bcast(torch.zeros(b, dtype=torch.int32, device=dev))
            seq_starts = _bcast(torch.zeros(b, dtype=torch.int32, device=dev))
            model.decode(tokens, positions, caches, slots, bt, seq_lens,
                         seq_starts=seq_starts)
