"""Serving engine: paged KV cache + continuous batching.

Native replacement for the reference's external basaran server image
(SURVEY.md §2b "server image"): greedy/temperature decode over a paged
KV cache, sized for the MI355X's 288 GB HBM3E (the cache pool is
allocated from measured free memory, not a guess), TP across up to 8
GPUs over xGMI.

Scheduling: one prefill (whole prompt in one pass) is admitted per step
when capacity allows, then all running sequences decode as one batch —
decode batches hit the gfx950 paged-decode kernel
(ops/csrc/attention_decode.hip).

Sliding-window attention (mistral, cfg.sliding_window): enforced here as
pure KV bookkeeping — RoPE is applied at absolute positions when keys
are written, so attention over any suffix of the cache is exact, and the
kernel needs no window parameter. Whole front blocks that fall entirely
outside the window are freed (bounded cache memory per sequence) and
the block table / seq_len presented to the kernel cover only the
retained suffix. Granularity is BLOCK_SIZE: a query attends to between
W and W+BLOCK_SIZE-1 trailing tokens; prompts longer than W prefill
with full causal attention (exact for prompts <= W, which covers the
supported max prompt for mistral-7b).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

import torch

from .. import ops
from ..models import Transformer, build_model
from ..parallel import comm
from ..utils.trace import get_tracer


class BlockAllocator:
    """Free-list allocator with reference counts: prefix-cached blocks
    are shared across requests (and held by the cache itself); a block
    returns to the free list when its last reference is released."""

    def __init__(self, num_blocks: int):
        self.free = list(range(num_blocks - 1, -1, -1))
        self.num_blocks = num_blocks
        self.refs = [0] * num_blocks

    def alloc(self, n: int) -> list[int]:
        if n > len(self.free):
            raise RuntimeError("KV cache exhausted")
        out = [self.free.pop() for _ in range(n)]
        for b in out:
            self.refs[b] = 1
        return out

    def share(self, block: int) -> None:
        assert self.refs[block] > 0, "sharing a free block"
        self.refs[block] += 1

    def release(self, blocks: list[int]):
        for b in blocks:
            self.refs[b] -= 1
            assert self.refs[b] >= 0, "double release"
            if self.refs[b] == 0:
                self.free.append(b)


@dataclass
class Request:
    request_id: int
    prompt_ids: list[int]
    max_new_tokens: int = 64
    temperature: float = 0.0
    top_p: float = 1.0
    logprobs: int | None = None     # record top-k logprobs per token
    seed: int | None = None         # per-request sampling seed override
    presence_penalty: float = 0.0   # subtract once per seen token
    frequency_penalty: float = 0.0  # subtract per occurrence
    repetition_penalty: float = 1.0  # HF-style multiplicative penalty
    stop_ids: tuple[int, ...] = ()
    # state
    output_ids: list[int] = field(default_factory=list)
    # per output token, when logprobs is set: {"logprob": f, "top": [(id, f)]}
    logprob_data: list = field(default_factory=list)
    blocks: list[int] = field(default_factory=list)
    # tokens whose KV blocks were freed by the sliding window (always a
    # multiple of BLOCK_SIZE; blocks[] maps retained positions only)
    dropped: int = 0
    finished: bool = False
    created: float = field(default_factory=time.time)

    @property
    def seq_len(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)


class Engine:
    def __init__(self, model: Transformer | str, device=None,
                 dtype=torch.bfloat16, kv_blocks: int | None = None,
                 max_batch: int = 64, mem_fraction: float = 0.85, seed: int = 0,
                 load_in_8bit: bool = False, prefix_cache: bool | None = None,
                 kv_fp8: bool | None = None):
        self.device = device if device is not None else (
            f"cuda:{comm.local_rank()}" if torch.cuda.is_available() else "cpu")
        if isinstance(model, str):
            model = build_model(model, dtype=dtype, seed=seed,
                                device=self.device)
        self.model = model.to(self.device).eval()
        if torch.cuda.is_available():
            # TP>1 included: each rank fuses ITS OWN qkv / gate-up weight
            # shards and registers decode-GEMM layouts for them — the
            # collective order (o/down all-reduce per layer) is untouched,
            # so the worker-follow protocol stays aligned.
            from ..models.transformer import fuse_for_inference
            fuse_for_inference(self.model,
                               load_in_8bit=load_in_8bit and model.tp == 1)
        if load_in_8bit and model.tp > 1:
            print("engine: MODEL_LOAD_IN_8BIT ignored at TP>1 "
                  "(fp8 decode path is single-GPU; 288 GB/GPU rarely "
                  "needs 8-bit at TP>1)", flush=True)
        self.cfg = model.cfg
        if torch.cuda.is_available() and model.cfg.head_dim not in (64, 128,
                                                                    256):
            raise NotImplementedError(
                f"paged decode supports head_dim 64/128/256 on gfx950; "
                f"{model.cfg.name} has {model.cfg.head_dim}")
        self.bs = ops.BLOCK_SIZE
        self.max_batch = max_batch
        # Admission budget per step: the default 1 matches the classic
        # one-prefill-then-decode cadence (decode latency bounded); the
        # HTTP server raises it for bursty arrival patterns where
        # admission throughput dominates (scripts/bench_http.py).
        self.max_prefills_per_step = 1
        self.seed = seed
        # fp8-e4m3 KV cache (RB_KV_FP8=1 or kv_fp8=True): halves the
        # decode-attention read stream and doubles KV capacity — the
        # long-context lever for the 288 GB/GPU KV story (BASELINE #5).
        import os as _os0
        self.kv_fp8 = (kv_fp8 if kv_fp8 is not None
                       else _os0.environ.get("RB_KV_FP8", "0") == "1")
        if kv_blocks is None:
            kv_blocks = self._auto_kv_blocks(mem_fraction)
        if comm.is_dist() and comm.world_size() > 1 and model.tp > 1:
            # Every rank must agree on the pool size: rank 0's scheduler
            # hands out slot indices that worker ranks index their OWN
            # caches with — a rank with less free memory would otherwise
            # read/write out of bounds. Take the fleet minimum.
            t = torch.tensor([kv_blocks], dtype=torch.int64,
                             device=self.device if
                             torch.cuda.is_available() else "cpu")
            torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MIN)
            kv_blocks = int(t.item())
        # last block is reserved as the dummy block hipGraph-padded decode
        # rows read/write (serve/graph.py)
        self.allocator = BlockAllocator(kv_blocks - 1)
        self.caches = self.model.alloc_caches(kv_blocks, self.device,
                                              fp8=self.kv_fp8)
        self.dummy_block = kv_blocks - 1
        self.waiting: list[Request] = []
        self.running: list[Request] = []
        self._next_id = 0
        # sliding window bounds per-seq blocks at W//bs + 2 (retained
        # suffix <= W + bs tokens incl. the one being appended); prompts
        # prefill in full but front blocks are dropped before the first
        # decode, so this is the true block-table width for decode.
        seq_cap = self.cfg.max_seq_len
        if self.cfg.sliding_window:
            seq_cap = min(seq_cap, self.cfg.sliding_window + 2 * self.bs)
        # +1: _admit allocates ceil(S/bs)+1 blocks (prompt + headroom for
        # the first decoded token), so a near-max-length prompt can hold
        # ceil(seq_cap/bs)+1 blocks — the graph staging table must be at
        # least that wide or btn[i, :len(row)] raises mid-flight.
        self.max_blocks_per_seq = min(
            kv_blocks, (seq_cap + self.bs - 1) // self.bs + 1)
        self._graphed = None
        # hipGraphs drive the single-GPU decode; the TP>1 path runs eager
        # so the worker-follow protocol (serve/tp_worker.py) sees every
        # collective.
        self.tp = comm.world_size()
        self.use_graphs = torch.cuda.is_available() and self.tp == 1
        # Prefix caching (opt-in, RB_PREFIX_CACHE=1 or prefix_cache=True):
        # full 16-token prompt-prefix chunks are shared across requests by
        # refcount — a hit skips the KV writes for those chunks (the
        # prefill still computes the full prompt, so attention inputs are
        # identical; the kernels are untouched, only the block table
        # changes). Bounds cache memory for common-system-prompt serving.
        import os as _os
        # default-on since the GPU validation pass (r2: shared-prompt
        # decode parity vs an uncached engine); RB_PREFIX_CACHE=0 disables
        self.prefix_cache_enabled = (
            prefix_cache if prefix_cache is not None
            else _os.environ.get("RB_PREFIX_CACHE", "1") == "1")
        from collections import OrderedDict
        self._pc: "OrderedDict[tuple, int]" = OrderedDict()
        self.stats = {"steps": 0, "prefills": 0, "decode_tokens": 0,
                      "preemptions": 0, "prefix_hits": 0,
                      "prefix_hit_blocks": 0, "window_dropped_blocks": 0}

    def _auto_kv_blocks(self, mem_fraction: float) -> int:
        elem = (self.cfg.head_dim + 16) if self.kv_fp8 \
            else self.cfg.head_dim * 2
        bytes_per_block = (2 * self.cfg.num_layers * self.model.local_kv_heads()
                           * self.bs * elem)
        if torch.cuda.is_available():
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * mem_fraction) - (2 << 30)  # activations headroom
        else:
            budget = 64 << 20
        return max(16, budget // bytes_per_block)

    # -- request API ------------------------------------------------------------
    def submit(self, prompt_ids: list[int], max_new_tokens: int = 64,
               temperature: float = 0.0, top_p: float = 1.0,
               logprobs: int | None = None, seed: int | None = None,
               presence_penalty: float = 0.0,
               frequency_penalty: float = 0.0,
               repetition_penalty: float = 1.0,
               stop_token_ids: tuple[int, ...] = ()) -> Request:
        req = Request(self._next_id, list(prompt_ids), max_new_tokens,
                      temperature, top_p, logprobs, seed,
                      presence_penalty, frequency_penalty,
                      repetition_penalty, tuple(stop_token_ids))
        self._next_id += 1
        self.waiting.append(req)
        return req

    @staticmethod
    def _apply_penalties(row, r: Request):
        """OpenAI presence/frequency penalties over the text so far
        (prompt + generated). Torch ops on the logits row, outside the
        decode hipGraph — zero cost for requests that don't ask."""
        if not (r.presence_penalty or r.frequency_penalty or
                r.repetition_penalty != 1.0):
            return row
        from collections import Counter
        cnt = Counter(r.prompt_ids)
        cnt.update(r.output_ids)
        ids = torch.tensor(list(cnt.keys()), dtype=torch.long,
                           device=row.device)
        c = torch.tensor([float(v) for v in cnt.values()],
                         device=row.device)
        row = row.clone().float()
        row[ids] -= r.frequency_penalty * c + r.presence_penalty
        if r.repetition_penalty != 1.0:
            seen = row[ids]
            row[ids] = torch.where(seen > 0, seen / r.repetition_penalty,
                                   seen * r.repetition_penalty)
        return row

    @staticmethod
    def _record_logprobs(req: Request, logits_row, tok_id: int) -> None:
        """Top-k logprob bookkeeping for one sampled token (host sync;
        only runs for requests that asked — sampling already lives
        outside the hipGraph, so the graphed decode path is untouched)."""
        lp = torch.log_softmax(logits_row.float(), dim=-1)
        entry = {"logprob": float(lp[tok_id])}
        k = req.logprobs or 0
        if k > 0:
            v, idx = lp.topk(min(k, lp.numel()))
            entry["top"] = list(zip(idx.tolist(), v.tolist()))
        req.logprob_data.append(entry)

    def _admit(self) -> Request | None:
        if not self.waiting or len(self.running) >= self.max_batch:
            return None
        req = self.waiting[0]
        S = len(req.prompt_ids)
        hits: list[tuple[tuple, int]] = []
        if self.prefix_cache_enabled:
            for k in range(1, S // self.bs + 1):
                key = tuple(req.prompt_ids[:k * self.bs])
                b = self._pc.get(key)
                if b is None:
                    break
                hits.append((key, b))
        need = (S + self.bs - 1) // self.bs + 1 - len(hits)
        if need > len(self.allocator.free) and self.prefix_cache_enabled:
            self._evict_prefix(need - len(self.allocator.free),
                               protect={b for _, b in hits})
        if need > len(self.allocator.free):
            return None
        self.waiting.pop(0)
        if hits:
            self.stats["prefix_hits"] += 1
            self.stats["prefix_hit_blocks"] += len(hits)
        for key, b in hits:
            self.allocator.share(b)
            self._pc.move_to_end(key)
        req.blocks = [b for _, b in hits] + self.allocator.alloc(need)
        req._shared_chunks = len(hits)
        return req

    def _evict_prefix(self, n_blocks: int, protect: set = frozenset()):
        """Drop LRU prefix-cache entries until ~n_blocks come free (an
        entry only frees its block if no live request still shares it)."""
        for key in list(self._pc):
            if n_blocks <= 0:
                return
            b = self._pc[key]
            if b in protect:
                continue
            del self._pc[key]
            before = len(self.allocator.free)
            self.allocator.release([b])
            n_blocks -= len(self.allocator.free) - before

    def _pc_insert(self, req: Request) -> None:
        """After a successful prefill, publish the request's full prompt
        chunks (cache holds its own reference per block)."""
        S = len(req.prompt_ids)
        for k in range(1, S // self.bs + 1):
            key = tuple(req.prompt_ids[:k * self.bs])
            if key not in self._pc:
                b = req.blocks[k - 1]
                self._pc[key] = b
                self.allocator.share(b)
            self._pc.move_to_end(key)

    def flush_prefix_cache(self) -> None:
        for b in self._pc.values():
            self.allocator.release([b])
        self._pc.clear()

    # -- model invocations ------------------------------------------------------
    def _prefill(self, req: Request) -> int:
        S = len(req.prompt_ids)
        tokens = torch.tensor([req.prompt_ids], dtype=torch.long, device=self.device)
        positions = torch.arange(S, dtype=torch.int32, device=self.device)
        # prefix-cache hits: those chunks' KV already sit in the shared
        # blocks, so their writes are redirected to the dummy block
        # (trash); everything else lands in the request's own blocks.
        shared_tokens = getattr(req, "_shared_chunks", 0) * self.bs
        slots = torch.tensor(
            [(self.dummy_block * self.bs + i % self.bs) if i < shared_tokens
             else req.blocks[i // self.bs] * self.bs + i % self.bs
             for i in range(S)],
            dtype=torch.int32, device=self.device)
        if self.tp > 1:
            from .tp_worker import broadcast_prefill
            broadcast_prefill(tokens, positions, slots)
        logits = self.model.prefill(tokens, positions, self.caches, slots)
        # a per-request seed makes sampling reproducible across runs and
        # independent of the request_id the scheduler happened to assign
        base = (req.seed * 2654435761 if req.seed is not None
                else self.seed + req.request_id * 65537)
        logits0 = self._apply_penalties(logits[0], req)
        tok = ops.sample_tokens(logits0.unsqueeze(0), req.temperature,
                                top_p=req.top_p, seed=base + S)
        t = int(tok[0])
        if req.logprobs is not None:
            self._record_logprobs(req, logits0, t)
        return t

    def _decode_batch(self, reqs: list[Request]) -> list[int]:
        B = len(reqs)
        dev = self.device
        last = [r.prompt_ids[-1] if not r.output_ids else r.output_ids[-1]
                for r in reqs]
        # RoPE position of the token being generated = absolute seq_len - 1
        # (keys were roped at absolute positions too, so windowed suffixes
        # stay exact). Its kv slot / the attended seq_len use the VIRTUAL
        # position within the retained blocks: absolute minus dropped.
        pos = [r.seq_len - 1 for r in reqs]
        vpos = [r.seq_len - 1 - r.dropped for r in reqs]
        slots = []
        for r, v in zip(reqs, vpos):
            blk = r.blocks[v // self.bs]
            slots.append(blk * self.bs + v % self.bs)
        # strict sliding window: attend to exactly the last W virtual
        # positions (block dropping keeps memory bounded; seq_starts
        # removes the residual 0..15-token block slack)
        W = self.cfg.sliding_window
        starts = [max(0, v + 1 - W) if W else 0 for v in vpos]

        if self.use_graphs and B <= self.max_batch:
            if self._graphed is None:
                from .graph import GraphedDecoder
                self._graphed = GraphedDecoder(
                    self.model, self.caches, self.max_batch,
                    self.max_blocks_per_seq, self.dummy_block, dev)
            logits = self._graphed.decode(last, pos, slots,
                                          [r.blocks for r in reqs],
                                          [v + 1 for v in vpos], starts)
        else:
            maxb = max(len(r.blocks) for r in reqs)
            bt = torch.zeros(B, maxb, dtype=torch.int32)
            for i, r in enumerate(reqs):
                bt[i, :len(r.blocks)] = torch.tensor(r.blocks,
                                                     dtype=torch.int32)
            tokens = torch.tensor(last, dtype=torch.long, device=dev)
            positions = torch.tensor(pos, dtype=torch.int32, device=dev)
            slot_t = torch.tensor(slots, dtype=torch.int32, device=dev)
            seq_lens = torch.tensor([v + 1 for v in vpos], dtype=torch.int32,
                                    device=dev)
            start_t = torch.tensor(starts, dtype=torch.int32, device=dev)
            bt = bt.to(dev)
            if self.tp > 1:
                from .tp_worker import broadcast_decode
                broadcast_decode(tokens, positions, slot_t, bt, seq_lens,
                                 start_t)
            logits = self.model.decode(tokens, positions, self.caches, slot_t,
                                       bt, seq_lens, seq_starts=start_t)
        return self._sample_batch(logits, reqs)

    def _sample_batch(self, logits, reqs: list[Request]) -> list[int]:
        seed = self.seed + 1_000_003 * reqs[0].seq_len
        params = {(r.temperature, r.top_p) for r in reqs}
        plain = all(r.seed is None and not r.presence_penalty and
                    not r.frequency_penalty and
                    r.repetition_penalty == 1.0 for r in reqs)
        if len(params) == 1 and plain:
            t, p = params.pop()
            toks = ops.sample_tokens(logits, t, top_p=p, seed=seed)
            out = [int(x) for x in toks]
        else:
            # heterogeneous sampling params / custom seeds: row-by-row
            out = []
            for i, r in enumerate(reqs):
                rs = (r.seed * 2654435761 + r.seq_len
                      if r.seed is not None else seed + r.request_id)
                row = self._apply_penalties(logits[i], r)
                tok = ops.sample_tokens(row.unsqueeze(0), r.temperature,
                                        top_p=r.top_p, seed=rs)
                out.append(int(tok[0]))
        for i, r in enumerate(reqs):
            if r.logprobs is not None:
                self._record_logprobs(r, logits[i], out[i])
        return out

    def _apply_window(self) -> None:
        """Free whole front KV blocks that fall entirely outside the
        sliding window of the next query (absolute position seq_len - 1).
        Dropping whole blocks keeps the within-block slot offsets of the
        retained tokens unchanged, so no cache data moves."""
        w = self.cfg.sliding_window
        if not w:
            return
        for r in self.running:
            while r.dropped + self.bs <= r.seq_len - w and len(r.blocks) > 1:
                self.allocator.release([r.blocks.pop(0)])
                r.dropped += self.bs
                self.stats["window_dropped_blocks"] += 1

    def _preempt(self, r: Request) -> None:
        """Release a running request's cache and requeue it: its generated
        tokens become part of the prompt for the re-prefill."""
        self.allocator.release(r.blocks)
        r.blocks = []
        r.dropped = 0
        r.prompt_ids = r.prompt_ids + r.output_ids
        r.max_new_tokens -= len(r.output_ids)
        r.output_ids = []
        self.stats["preemptions"] += 1
        if hasattr(r, "_watch_sent"):
            r._watch_sent = 0  # post-preempt tokens are all new to watchers
        self.running.remove(r)
        self.waiting.insert(0, r)

    def _preempt_newest(self, exclude: Request) -> bool:
        for r in reversed(self.running):
            if r is not exclude:
                self._preempt(r)
                return True
        return False

    def cancel(self, request_id: int) -> bool:
        """Stop a request: waiting -> dropped now; running -> finishes at
        the next scheduler sweep."""
        for r in self.waiting:
            if r.request_id == request_id:
                self.waiting.remove(r)
                r.finished = True
                return True
        for r in self.running:
            if r.request_id == request_id:
                r.max_new_tokens = max(1, len(r.output_ids))
                return True
        return False

    # -- scheduler step ----------------------------------------------------------
    def step(self) -> list[Request]:
        """One engine iteration. Returns requests finished this step."""
        self.stats["steps"] += 1
        finished = []
        # sweep first: requests already satisfied (cancel() capped their
        # max_new_tokens between steps) must not decode one extra token
        self._sweep_finished(finished)
        req = self._admit()
        if req is None and not self.running and self.waiting:
            head = self.waiting[0]
            need = (len(head.prompt_ids) + self.bs - 1) // self.bs + 1
            if self.prefix_cache_enabled:
                for k in range(1, len(head.prompt_ids) // self.bs + 1):
                    if tuple(head.prompt_ids[:k * self.bs]) in self._pc:
                        need -= 1
                    else:
                        break
            if need > self.allocator.num_blocks:
                r = self.waiting.pop(0)
                r.finished = True
                raise RuntimeError(
                    f"prompt of {len(r.prompt_ids)} tokens cannot fit the KV "
                    f"cache ({self.allocator.num_blocks} blocks)")
        tr = get_tracer()
        n_admitted = 0
        while req is not None:
            try:
                if tr:
                    with tr.span("prefill", tokens=len(req.prompt_ids),
                                 request=req.request_id):
                        first = self._prefill(req)
                else:
                    first = self._prefill(req)
            except Exception:
                # _admit already popped req from waiting and allocated its
                # blocks; releasing here (shared prefix blocks are
                # refcounted, release decrements) keeps a failing request
                # from leaking KV pool forever.
                self.allocator.release(req.blocks)
                req.blocks = []
                req.finished = True
                raise
            self.stats["prefills"] += 1
            if self.prefix_cache_enabled:
                self._pc_insert(req)
            req.output_ids.append(first)
            self.running.append(req)
            # the prefill token may already satisfy the request
            self._sweep_finished(finished)
            n_admitted += 1
            req = (self._admit()
                   if n_admitted < self.max_prefills_per_step else None)
        # decode every step — admissions must not starve running
        # sequences (a freshly prefilled request decodes its second
        # token in the same step it was admitted)
        if self.running:
            self._apply_window()
            # ensure every running seq has a block for the next position;
            # when the pool is exhausted, preempt the newest sequence
            # (its blocks are freed and it requeues for a fresh prefill)
            # instead of failing the whole batch.
            for r in list(self.running):
                if r not in self.running:
                    continue
                while r.seq_len - r.dropped >= len(r.blocks) * self.bs:
                    if not self.allocator.free and self.prefix_cache_enabled:
                        # idle cached prefix blocks are reclaimable memory:
                        # evict before resorting to preemption
                        self._evict_prefix(1)
                    if self.allocator.free:
                        r.blocks.extend(self.allocator.alloc(1))
                    elif not self._preempt_newest(exclude=r):
                        # nothing left to preempt: preempt r itself
                        self._preempt(r)
                        break
            if self.running:
                if tr:
                    with tr.span("decode", batch=len(self.running)):
                        toks = self._decode_batch(self.running)
                else:
                    toks = self._decode_batch(self.running)
                self.stats["decode_tokens"] += len(toks)
                for r, t in zip(self.running, toks):
                    r.output_ids.append(t)
        self._sweep_finished(finished)
        return finished

    def _sweep_finished(self, finished: list[Request]) -> None:
        for r in list(self.running):
            if (len(r.output_ids) >= r.max_new_tokens
                    or (r.stop_ids and r.output_ids[-1] in r.stop_ids)
                    or r.seq_len >= self.cfg.max_seq_len):
                r.finished = True
                self.allocator.release(r.blocks)
                r.blocks = []
                self.running.remove(r)
                finished.append(r)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    # -- convenience --------------------------------------------------------------
    def generate(self, prompt_ids: list[int], max_new_tokens: int = 64,
                 temperature: float = 0.0) -> list[int]:
        req = self.submit(prompt_ids, max_new_tokens, temperature)
        while not req.finished:
            self.step()
        return req.output_ids
