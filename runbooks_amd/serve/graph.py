"""hipGraph-captured decode step.

The decode inner loop is launch-bound: ~8 small kernels x num_layers per
step plus five host->device tensor builds. This wraps the whole
model.decode(...) call in a hipGraph (torch.cuda.CUDAGraph is hipGraph on
ROCm) per batch-size bucket: static device input buffers are filled from
pinned staging with async copies, then one graph replay launches the
whole step (graph-replay floor ~10-16 us vs ~3.5 us/launch eager,
MI355X_MICROARCH.md price list).

Batch sizes are bucketed to powers of two; short batches pad with a
dummy sequence that reads/writes a reserved KV block, so replay shapes
stay fixed while requests come and go.
"""
from __future__ import annotations

import torch


def fixed_nsplit(batch: int, hkv: int, mfma: bool = False) -> int:
    """Work-split for the paged-decode kernel, chosen per bucket at capture
    time (the eager heuristic in ops/attention.py reads seq_lens.max() —
    a host sync, impossible inside a graph). base = batch*hkv workgroups
    per split; split until ~4x256 CUs are covered — the head-split GQA
    kernels run 4-5 waves/SIMD, so ~4 WGs/CU (1024 WGs) fills the
    machine (measured sweep: llama2-70b B=32 len-2048 1.75 vs 0.67
    TB/s, falcon MQA 0.72 vs 0.23, B=8 0.99-1.24 vs 0.16 —
    profiles/decode_attn_pipeline.md)."""
    base = max(1, batch * hkv)
    # MFMA path (transposed-V caches): per-wave work is block-granular
    # and latency-light — ~512 WGs is the knee (measured: llama2-70b
    # B=32 len-2048 best at nsplit 2, B=8 at 4-8); the scalar kernels
    # want ~1024.
    target = 512 if mfma else 1024
    if base >= target:
        return 1
    return min(16, max(1, target // base))


class GraphedDecoder:
    def __init__(self, model, caches, max_batch: int, max_blocks: int,
                 dummy_block: int, device):
        self.model = model
        self.caches = caches
        self.device = device
        self.max_blocks = max_blocks
        self.dummy_block = dummy_block
        self.buckets = []
        b = 1
        while b < max_batch:
            self.buckets.append(b)
            b *= 2
        self.buckets.append(max_batch)

        B = max_batch
        dev = device
        self.tokens = torch.zeros(B, dtype=torch.long, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.slots = torch.full((B,), dummy_block * _bs(), dtype=torch.int32,
                                device=dev)
        self.block_tables = torch.full((B, max_blocks), dummy_block,
                                       dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.seq_starts = torch.zeros(B, dtype=torch.int32, device=dev)
        pin = dict(dtype=torch.int64,
                   pin_memory=torch.cuda.is_available())
        self.h_staging = torch.zeros(B * 5 + B * max_blocks, **pin)
        self._np = self.h_staging.numpy()
        self.graphs = {}     # bucket -> (CUDAGraph, logits_out)
        self._pool = None

    # -- capture -----------------------------------------------------------
    def _capture(self, b: int):
        hkv = self.model.local_kv_heads()
        from ..ops.attention import _is_vt
        nsplit = fixed_nsplit(b, hkv,
                              mfma=_is_vt(*self.caches[0][:2]))
        args = (self.tokens[:b], self.positions[:b], self.caches,
                self.slots[:b], self.block_tables[:b], self.seq_lens[:b])
        kw = dict(nsplit=nsplit, seq_starts=self.seq_starts[:b])
        # warmup outside the graph (allocator + autotune settle)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.decode(*args, **kw)
        torch.cuda.current_stream().wait_stream(s)

        g = torch.cuda.CUDAGraph()
        if self._pool is None:
            with torch.cuda.graph(g):
                out = self.model.decode(*args, **kw)
            self._pool = g.pool()
        else:
            with torch.cuda.graph(g, pool=self._pool):
                out = self.model.decode(*args, **kw)
        self.graphs[b] = (g, out)

    # -- replay ------------------------------------------------------------
    def decode(self, tokens, positions, slots, block_rows, seq_lens,
               seq_starts=None):
        """All args host lists; block_rows is a list of per-seq block lists.
        Returns logits [len(tokens), V] (a view into the static output)."""
        n = len(tokens)
        b = next(x for x in self.buckets if x >= n)
        if b not in self.graphs:
            self._capture(b)
        g, out = self.graphs[b]
        bt = self._stage(b, tokens, positions, slots, block_rows, seq_lens,
                         seq_starts or [0] * n)

        h = self.h_staging
        self.tokens[:b].copy_(h[0:b], non_blocking=True)
        self.positions[:b].copy_(h[b:2 * b], non_blocking=True)
        self.slots[:b].copy_(h[2 * b:3 * b], non_blocking=True)
        self.seq_lens[:b].copy_(h[3 * b:4 * b], non_blocking=True)
        self.seq_starts[:b].copy_(h[4 * b:5 * b], non_blocking=True)
        self.block_tables[:b].copy_(bt, non_blocking=True)
        g.replay()
        return out[:n]

    def _stage(self, b, tokens, positions, slots, block_rows, seq_lens,
               seq_starts):
        """Fill the pinned staging buffer for bucket size b (numpy view:
        C-speed fills instead of one small torch.tensor per row). Unit-
        tested on CPU (the rest of this class needs a GPU)."""
        n = len(tokens)
        mb = self.max_blocks
        hn = self._np
        hn[0:n] = tokens
        hn[b:b + n] = positions
        hn[2 * b:2 * b + n] = slots
        hn[3 * b:3 * b + n] = seq_lens
        hn[4 * b:4 * b + n] = seq_starts
        btn = hn[5 * b:5 * b + b * mb].reshape(b, mb)
        btn.fill(self.dummy_block)
        for i, row in enumerate(block_rows):
            btn[i, :len(row)] = row
        # pad rows beyond n: dummy sequence at position 0, length 1
        if n < b:
            hn[n:b] = 0
            hn[b + n:2 * b] = 0
            hn[2 * b + n:3 * b] = self.dummy_block * _bs()
            hn[3 * b + n:4 * b] = 1
            hn[4 * b + n:5 * b] = 0
        return self.h_staging[5 * b:5 * b + b * mb].view(b, mb)


def _bs() -> int:
    from .. import ops
    return ops.BLOCK_SIZE
