"""Model-family + engine + LoRA behavior on CPU (fp32 references)."""
import pytest
import torch

from runbooks_amd.models import build_model, get_config, list_configs
from runbooks_amd.serve import Engine
from runbooks_amd.train import apply_lora, lora_state_dict, merge_lora


@pytest.mark.parametrize("name", ["tiny-llama", "tiny-falcon", "tiny-opt"])
def test_forward_shapes(name):
    m = build_model(name, dtype=torch.float32)
    out = m(torch.randint(0, 256, (2, 16)))
    assert out.shape == (2, 16, m.cfg.vocab_size)
    assert torch.isfinite(out).all()


def test_registry_has_flagship_configs():
    for name in ("llama2-7b", "llama2-70b", "llama3-8b", "llama3-70b",
                 "mistral-7b", "falcon-40b", "falcon-7b", "opt-125m"):
        cfg = get_config(name)
        assert cfg.num_heads % cfg.num_kv_heads == 0
    # llama2-7b parameter count sanity (±10%)
    assert abs(get_config("llama2-7b").params_b - 6.7) < 0.7
    assert abs(get_config("llama2-70b").params_b - 69) < 7
    assert abs(get_config("opt-125m").params_b - 0.125) < 0.05


@pytest.mark.parametrize("name", ["tiny-llama", "tiny-falcon", "tiny-opt"])
def test_engine_matches_full_forward(name):
    """Paged prefill+decode must equal the no-cache forward (greedy)."""
    m = build_model(name, dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    prompt = [5, 9, 2, 7, 1]
    out = eng.generate(list(prompt), max_new_tokens=6)
    seq = list(prompt)
    for _ in range(6):
        logits = m(torch.tensor([seq]))
        seq.append(int(logits[0, -1].argmax()))
    assert out == seq[len(prompt):], f"{name}: {out} != {seq[len(prompt):]}"


def test_engine_batched_decode_matches_serial():
    m = build_model("tiny-llama", dtype=torch.float32)
    prompts = [[1, 2, 3], [9, 8, 7, 6], [4, 4]]
    serial = []
    for p in prompts:
        eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
        serial.append(eng.generate(list(p), max_new_tokens=4))
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, max_batch=4)
    reqs = [eng.submit(list(p), max_new_tokens=4) for p in prompts]
    while eng.has_work():
        eng.step()
    for r, s in zip(reqs, serial):
        assert r.output_ids == s


def test_lora_freezes_base_and_merges():
    torch.manual_seed(0)
    m = build_model("tiny-llama", dtype=torch.float32)
    wrapped = apply_lora(m, r=4, alpha=8)
    assert len(wrapped) > 0
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert all("lora_" in n for n in trainable)
    assert len(lora_state_dict(m)) == 2 * len(wrapped)

    x = torch.randint(0, 256, (1, 8))
    before = m(x)
    # train the adapters a little so B != 0
    opt = torch.optim.SGD([p for p in m.parameters() if p.requires_grad], lr=0.1)
    loss = m(x).square().mean()
    loss.backward()
    opt.step()
    after_train = m(x)
    assert not torch.allclose(before, after_train)
    merge_lora(m)
    merged = m(x)
    assert torch.allclose(after_train, merged, atol=1e-4)


def test_kv_cache_exhaustion_raises():
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=2)
    with pytest.raises(RuntimeError):
        eng.generate(list(range(40)), max_new_tokens=4)


def test_block_allocator_release():
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    free0 = len(eng.allocator.free)
    eng.generate([1, 2, 3], max_new_tokens=3)
    assert len(eng.allocator.free) == free0, "blocks leaked"


def test_fuse_for_inference_matches_unfused():
    """Fused QKV / gate-up serving path == separate projections."""
    import torch
    from runbooks_amd.models import build_model
    from runbooks_amd.models.transformer import fuse_for_inference

    m = build_model("tiny-llama", dtype=torch.float32, tp=1, seed=5)
    tokens = torch.randint(0, 256, (2, 9))
    with torch.no_grad():
        ref = m(tokens)
    fuse_for_inference(m)
    with torch.no_grad():
        got = m(tokens)
    assert torch.allclose(got, ref, atol=1e-6)
    # parameters are views into the fused tensors (no weight duplication)
    blk = m.blocks[0]
    assert blk.attn.q_proj.weight.data_ptr() == blk.attn._qkv_w.data_ptr()


def test_kv_preemption_under_pressure():
    """When the block pool runs dry mid-decode, the newest sequence is
    preempted (requeued for re-prefill) instead of crashing the batch,
    and everyone still finishes with the right token counts."""
    m = build_model("tiny-llama", dtype=torch.float32)
    # 8 usable blocks x 16 tokens; three growing sequences overflow it
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=9, seed=2)
    reqs = [eng.submit([i + 1] * 30, max_new_tokens=24) for i in range(3)]
    for _ in range(500):
        if not eng.has_work():
            break
        eng.step()
    assert not eng.has_work(), "scheduler wedged"
    for r in reqs:
        assert r.finished
    # total generated = 24 per request even across preemptions
    assert all(len(r.prompt_ids) + r.max_new_tokens == 30 + 24 or
               len(r.output_ids) == r.max_new_tokens for r in reqs)
    eng.flush_prefix_cache()   # cache refs are intentional retention
    assert len(eng.allocator.free) == eng.allocator.num_blocks


def test_preempted_output_preserved():
    """A preempted request's already-generated tokens are carried into the
    re-prefill prompt; the final combined output is contiguous."""
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=2)
    ref = eng.generate([7, 8, 9], max_new_tokens=10)

    eng2 = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=2)
    r = eng2.submit([7, 8, 9], max_new_tokens=10)
    for _ in range(4):
        eng2.step()
    eng2._preempt(r)  # force a mid-flight preemption
    while eng2.has_work():
        eng2.step()
    combined = r.prompt_ids[3:] + r.output_ids
    assert len(combined) == 10
    # greedy continuation after re-prefill matches the uninterrupted run
    assert combined == ref, (combined, ref)


@pytest.mark.parametrize("name", ["tiny-opt", "tiny-falcon", "tiny-gemma",
                                  "tiny-qwen", "tiny-gpt2"])
def test_engine_generates_all_families(name):
    """Decode path per family: OPT (learned positions, no rope, biases)
    and falcon (parallel residual, MQA) — the BASELINE config families."""
    m = build_model(name, dtype=torch.float32, seed=6)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=9)
    out = eng.generate([5, 4, 3], max_new_tokens=6)
    assert len(out) == 6
    assert all(0 <= t < m.cfg.vocab_size for t in out)
    # decode must agree with the full-forward argmax continuation
    full = [5, 4, 3]
    for _ in range(6):
        with torch.no_grad():
            logits = m(torch.tensor([full]))
        full.append(int(logits[0, -1].argmax()))
    assert out == full[3:], (out, full[3:])


def test_graphed_decoder_staging():
    """GraphedDecoder._stage fills the pinned buffer correctly, including
    dummy-padding of rows beyond the live batch (GPU-independent part of
    serve/graph.py)."""
    from runbooks_amd.serve.graph import GraphedDecoder

    gd = GraphedDecoder(model=None, caches=None, max_batch=8, max_blocks=4,
                        dummy_block=99, device="cpu")
    bt = gd._stage(4, tokens=[7, 8], positions=[3, 5], slots=[12, 20],
                   block_rows=[[1, 2], [4]], seq_lens=[4, 6],
                   seq_starts=[0, 2])
    h = gd.h_staging
    assert h[0:4].tolist() == [7, 8, 0, 0]          # tokens + pad
    assert h[4:8].tolist() == [3, 5, 0, 0]          # positions + pad
    assert h[8:10].tolist() == [12, 20]             # live slots
    assert (h[10:12] == 99 * 16).all()              # dummy slots
    assert h[12:16].tolist() == [4, 6, 1, 1]        # seq_lens + pad
    assert h[16:20].tolist() == [0, 2, 0, 0]        # seq_starts + pad
    assert bt.shape == (4, 4)
    assert bt[0].tolist() == [1, 2, 99, 99]
    assert bt[1].tolist() == [4, 99, 99, 99]
    assert (bt[2:] == 99).all()


@pytest.mark.parametrize("seed", [1234, 77])
def test_engine_scheduler_fuzz(seed):
    """Randomized scheduling traffic: submits, cancels and cache-pressure
    preemptions interleaved; invariants checked every step — no KV block
    is double-owned, nothing leaks, everything terminates."""
    import random

    rng = random.Random(seed)
    m = build_model("tiny-llama", dtype=torch.float32, seed=3)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=24,
                 max_batch=4, seed=5)
    live = []
    total_submitted = 0
    for step in range(400):
        if rng.random() < 0.25 and total_submitted < 25:
            plen = rng.randint(1, 40)
            r = eng.submit([rng.randrange(256) for _ in range(plen)],
                           max_new_tokens=rng.randint(1, 12))
            live.append(r)
            total_submitted += 1
        if rng.random() < 0.05 and live:
            eng.cancel(rng.choice(live).request_id)
        if eng.has_work():
            try:
                eng.step()
            except RuntimeError:
                pass  # oversize prompt rejected
        # invariants (refcount-aware: prefix cache shares blocks):
        # every reference is accounted for, free list = refs==0 exactly
        owned = [b for r in eng.running for b in r.blocks]
        assert all(0 <= b < eng.allocator.num_blocks for b in owned)
        from collections import Counter
        held = Counter(owned)
        held.update(eng._pc.values())
        for b, n in held.items():
            assert eng.allocator.refs[b] == n, f"refcount drift on {b}"
        live_set = set(held)
        assert not live_set & set(eng.allocator.free), "freed live block"
        assert len(eng.allocator.free) == \
            sum(1 for r in eng.allocator.refs if r == 0), "free-list drift"
        assert len(eng.allocator.free) + len(live_set) == \
            eng.allocator.num_blocks, "block leak"
    # drain
    for _ in range(3000):
        if not eng.has_work():
            break
        try:
            eng.step()
        except RuntimeError:
            pass
    assert not eng.has_work(), "scheduler did not terminate"
    eng.flush_prefix_cache()
    assert len(eng.allocator.free) == eng.allocator.num_blocks
    for r in live:
        assert r.finished
        assert len(r.output_ids) <= r.max_new_tokens or r.max_new_tokens <= 0


# --- sliding-window attention (mistral; engine-level KV bookkeeping) -------

def _windowed_model(window):
    import dataclasses
    cfg = dataclasses.replace(get_config("tiny-llama"), name="tiny-window",
                              sliding_window=window)
    return build_model(cfg, dtype=torch.float32)


def test_sliding_window_noop_below_window():
    """For sequences shorter than the window the windowed engine is
    bit-identical to the unwindowed one (no blocks dropped -> exact)."""
    torch.manual_seed(0)
    m_full = build_model("tiny-llama", dtype=torch.float32)
    torch.manual_seed(0)
    m_win = _windowed_model(64)
    eng_f = Engine(m_full, device="cpu", dtype=torch.float32, kv_blocks=64)
    eng_w = Engine(m_win, device="cpu", dtype=torch.float32, kv_blocks=64)
    prompt = [3, 1, 4, 1, 5]
    n = 20  # seq stays at 25 < 64
    assert eng_w.generate(list(prompt), n) == eng_f.generate(list(prompt), n)


def test_sliding_window_frees_blocks_and_bounds_table():
    """Long generation: front blocks get freed (bounded KV memory), the
    dropped count stays block-aligned, and the block table presented to
    decode never exceeds window//bs + 2 rows."""
    from runbooks_amd import ops
    w, bs = 32, ops.BLOCK_SIZE
    m = _windowed_model(w)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    seen = {"max_blocks": 0, "checked": 0}
    orig = eng._decode_batch

    def spy(reqs):
        for r in reqs:
            assert r.dropped % bs == 0
            retained = r.seq_len - r.dropped
            # retained context: within [w, w + bs) once past the window
            if r.seq_len > w + bs:
                assert w <= retained < w + bs, (r.seq_len, r.dropped)
                seen["checked"] += 1
            seen["max_blocks"] = max(seen["max_blocks"], len(r.blocks))
        return orig(reqs)

    eng._decode_batch = spy
    req = eng.submit([7] * 8, max_new_tokens=90)
    while eng.has_work():
        eng.step()
    assert req.finished and len(req.output_ids) == 90
    assert seen["checked"] > 0, "window never engaged"
    assert seen["max_blocks"] <= w // bs + 2
    # all blocks (incl. dropped ones) returned to the pool
    assert len(eng.allocator.free) == eng.allocator.num_blocks


def test_sliding_window_long_prompt_prefills_then_trims():
    """A prompt longer than the window prefills in full, then the first
    decode step trims the table down to the window."""
    from runbooks_amd import ops
    w = 32
    m = _windowed_model(w)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    req = eng.submit(list(range(1, 61)), max_new_tokens=5)  # 60-token prompt
    eng.step()  # prefill (full-causal) + first decode: window applied
    assert req.dropped > 0
    assert req.seq_len - req.dropped < w + ops.BLOCK_SIZE
    while eng.has_work():
        eng.step()
    assert len(req.output_ids) == 5


def test_sliding_window_matches_masked_forward():
    """Exactness of the window semantics: greedy windowed decode must
    equal a from-scratch forward under the equivalent banded attention
    mask. With a rolling KV cache, each token's K/V are computed ONCE
    under its own generation-time window (so deeper layers see a growing
    receptive field -- mistral's actual semantics; a fresh prefill of the
    retained suffix is NOT equivalent). The mask reference reproduces
    that exactly: prompt rows are fully causal (prefill), generated row
    i attends [block_drop(i), i] per the engine's block-aligned rule."""
    import math
    from runbooks_amd import ops
    w, bs = 32, ops.BLOCK_SIZE
    m = _windowed_model(w)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    prompt = [5, 9, 2, 7, 1, 8, 3, 6]
    req = eng.submit(list(prompt), max_new_tokens=56, temperature=0.0)
    while eng.has_work():
        eng.step()
    all_ids = req.prompt_ids + req.output_ids
    L, P = len(all_ids), len(prompt)

    mask = torch.zeros(L, L, dtype=torch.bool)
    for i in range(L):
        # strict W: generated row i attends exactly [i+1-w, i] (the
        # engine passes seq_starts to the decode kernel; block dropping
        # only bounds memory)
        start = 0 if i < P else max(0, i + 1 - w)
        mask[i, start:i + 1] = True

    def masked_attn(q, k, v, scale=None, q_block=256):
        B, S, Hq, Dh = q.shape
        g = Hq // k.shape[2]
        if g > 1:
            k, v = (t.repeat_interleave(g, dim=2) for t in (k, v))
        s = torch.einsum("bihd,bjhd->bhij", q.float(), k.float())
        s = (s * (scale or 1 / math.sqrt(Dh))).masked_fill(
            ~mask[:S, :S], float("-inf"))
        return torch.einsum("bhij,bjhd->bihd", s.softmax(-1),
                            v.float()).to(q.dtype)

    orig = ops.causal_attention
    ops.causal_attention = masked_attn
    try:
        with torch.no_grad():
            logits = m(torch.tensor([all_ids]))
    finally:
        ops.causal_attention = orig
    ref = logits[0, P - 1:L - 1].argmax(-1).tolist()
    assert req.output_ids == ref, (req.output_ids, ref)


# --- prefix caching (refcounted shared prompt blocks) ----------------------

def test_prefix_cache_shares_blocks_and_matches_uncached():
    """Identical prompt prefixes share KV blocks; outputs are identical
    to an uncached engine (the shared blocks hold bitwise-identical KV,
    written once by the first prefill)."""
    m = build_model("tiny-llama", dtype=torch.float32)
    base = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=3)
    pc = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=3,
                prefix_cache=True)
    sys_prompt = list(range(1, 33))  # 2 full chunks (bs=16)
    prompts = [sys_prompt + [40 + i] for i in range(3)]
    want = [base.generate(list(p), max_new_tokens=5) for p in prompts]

    reqs = [pc.submit(list(p), max_new_tokens=5) for p in prompts]
    shared = []
    while pc.has_work():
        pc.step()
        shared = [getattr(r, "_shared_chunks", 0) for r in reqs]
    assert [r.output_ids for r in reqs] == want
    # later requests hit the 2-chunk prefix
    assert any(s == 2 for s in shared[1:]), shared
    # both chunks cached exactly once; cache holds one ref each
    assert len(pc._pc) == 2
    free0 = len(pc.allocator.free)
    pc.flush_prefix_cache()
    assert len(pc.allocator.free) == free0 + 2
    assert len(pc.allocator.free) == pc.allocator.num_blocks


def test_prefix_cache_eviction_under_pressure():
    """Distinct prompts churn a small pool: LRU cache entries get
    evicted so admission never wedges, and accounting stays exact."""
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=10, seed=1,
                 prefix_cache=True)
    for i in range(6):
        out = eng.generate([i * 7 % 250 + 1] * 20, max_new_tokens=3)
        assert len(out) == 3
    # pool of 9 usable blocks, each prompt caches 1 chunk: evictions
    # must have kept cached+free == total
    cached = len(eng._pc)
    assert cached >= 1
    assert len(eng.allocator.free) + cached == eng.allocator.num_blocks
    eng.flush_prefix_cache()
    assert len(eng.allocator.free) == eng.allocator.num_blocks


def test_prefix_cache_with_sliding_window():
    """Prefix sharing composes with the rolling window: dropped shared
    blocks just decrement the refcount; outputs still match the
    uncached windowed engine."""
    m = _windowed_model(32)
    base = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=2)
    pc = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=2,
                prefix_cache=True)
    prompt = list(range(1, 33))
    want = base.generate(list(prompt), max_new_tokens=40)
    got1 = pc.generate(list(prompt), max_new_tokens=40)
    got2 = pc.generate(list(prompt), max_new_tokens=40)  # cache hit run
    assert got1 == want and got2 == want
    pc.flush_prefix_cache()
    assert len(pc.allocator.free) == pc.allocator.num_blocks


@pytest.mark.parametrize("seed", [99, 5])
def test_engine_fuzz_prefix_cache_and_window(seed):
    """Randomized traffic with prefix caching ON and a sliding window:
    per-block refcounts must equal (live request holders) + (cache
    holds), the free list must be exactly the zero-ref blocks, and
    everything terminates with a clean pool after a cache flush."""
    import random

    rng = random.Random(seed)
    m = _windowed_model(32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=28,
                 max_batch=4, seed=5, prefix_cache=True)
    # few shared prompt templates -> lots of prefix hits
    templates = [[i + 1] * 24 for i in range(3)]
    live = []
    submitted = 0
    for step in range(400):
        if rng.random() < 0.3 and submitted < 25:
            p = list(rng.choice(templates)) + \
                [rng.randrange(256) for _ in range(rng.randint(0, 8))]
            live.append(eng.submit(
                p, max_new_tokens=rng.randint(1, 30),
                temperature=rng.choice([0.0, 0.8]),
                logprobs=rng.choice([None, 0, 2]),
                seed=rng.choice([None, 7]),
                presence_penalty=rng.choice([0.0, 0.5]),
                frequency_penalty=rng.choice([0.0, 1.0])))
            submitted += 1
        if rng.random() < 0.05 and live:
            eng.cancel(rng.choice(live).request_id)
        if eng.has_work():
            eng.step()
        # refcount invariants
        counts = [0] * eng.allocator.num_blocks
        for r in eng.running:
            assert r.dropped % eng.bs == 0
            for b in r.blocks:
                counts[b] += 1
        for b in eng._pc.values():
            counts[b] += 1
        assert counts == eng.allocator.refs, (step, counts,
                                              eng.allocator.refs)
        free = sorted(eng.allocator.free)
        assert free == sorted(set(free)), "duplicate free blocks"
        assert free == [b for b in range(eng.allocator.num_blocks)
                        if counts[b] == 0]
    for _ in range(3000):
        if not eng.has_work():
            break
        eng.step()
    assert not eng.has_work()
    assert eng.stats["prefix_hits"] > 0, "fuzz never hit the cache"
    eng.flush_prefix_cache()
    assert len(eng.allocator.free) == eng.allocator.num_blocks
    for r in live:
        assert r.finished


def test_geglu_fused_path_matches_eager(monkeypatch):
    """RB_FUSED_GEGLU=1: gemma's gate/up fuse into one GEMM; the packed
    GeGLU epilogue (torch fallback on CPU; csrc geglu_packed on GPU)
    must equal the eager projections."""
    from runbooks_amd.models.transformer import fuse_for_inference

    m = build_model("tiny-gemma", dtype=torch.float32, seed=8)
    tokens = torch.randint(0, 256, (2, 9))
    with torch.no_grad():
        ref = m(tokens)
    monkeypatch.setenv("RB_FUSED_GEGLU", "1")
    fuse_for_inference(m)
    assert getattr(m.blocks[0].mlp, "_gateup_w", None) is not None
    with torch.no_grad():
        got = m(tokens)
    assert torch.allclose(got, ref, atol=1e-6)


def test_frequency_penalty_prevents_repetition():
    """A huge frequency penalty (greedy) forces every generated token to
    be distinct from all prior text; without it the tiny random model
    repeats (greedy cycles)."""
    m = build_model("tiny-llama", dtype=torch.float32, seed=1)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    plain = eng.generate([5, 5, 5], max_new_tokens=20)
    assert len(set(plain)) < 20, "expected the unpenalized run to repeat"
    r = eng.submit([5, 5, 5], max_new_tokens=20, frequency_penalty=100.0)
    while not r.finished:
        eng.step()
    out = r.output_ids
    assert len(set(out)) == 20, out
    assert 5 not in out  # prompt tokens penalized too (OpenAI semantics)


def test_rope_rotation_invariants():
    """RoPE reference invariants on CPU: per-pair norms are preserved
    (it's a rotation) and q·k depends only on the relative position —
    the property windowed/suffix attention exactness relies on."""
    from runbooks_amd import ops
    from runbooks_amd.models.transformer import Transformer  # noqa: F401

    m = build_model("tiny-llama", dtype=torch.float32)
    cos, sin = m.rope_cos, m.rope_sin
    torch.manual_seed(0)
    q = torch.randn(1, 4, 16)
    k = torch.randn(1, 4, 16)

    def rot(x, p):
        return ops.rope(x, cos, sin, torch.tensor([p], dtype=torch.int32))

    assert torch.allclose(rot(q, 7).norm(), q.norm(), atol=1e-5)
    # relative property: <R(p1)q, R(p2)k> == <R(p1+d)q, R(p2+d)k>
    def score(p1, p2):
        return torch.einsum("bhd,bhd->bh", rot(q, p1), rot(k, p2))

    assert torch.allclose(score(3, 1), score(23, 21), atol=1e-4)
    assert torch.allclose(score(10, 10), score(50, 50), atol=1e-4)
    assert not torch.allclose(score(3, 1), score(3, 2), atol=1e-3)


def test_stop_token_ids_finish_early():
    """Engine finishes a request the moment it emits a stop token id
    (the HTTP layer wires the tokenizer EOS here)."""
    m = build_model("tiny-llama", dtype=torch.float32, seed=3)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    plain = eng.generate([9, 9, 9], max_new_tokens=8)
    target = plain[2]
    r = eng.submit([9, 9, 9], max_new_tokens=8, stop_token_ids=(target,))
    while not r.finished:
        eng.step()
    assert r.output_ids == plain[:3], (r.output_ids, plain)


def test_repetition_penalty():
    """HF-style multiplicative penalty: a large value forbids repeats
    among positive-logit tokens (greedy)."""
    m = build_model("tiny-llama", dtype=torch.float32, seed=1)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    plain = eng.generate([4, 4, 4], max_new_tokens=16)
    assert len(set(plain)) < 16
    r = eng.submit([4, 4, 4], max_new_tokens=16, repetition_penalty=1e6)
    while not r.finished:
        eng.step()
    # tokens with positive logits get crushed after first use; repeats
    # can only come from the (rare) all-negative rows
    assert len(set(r.output_ids)) > len(set(plain))


def test_fp8_quant_roundtrip_cpu():
    """OCP e4m3 row-wise quantization on CPU (torch float8 dtype):
    round-trip error bounded by the format's relative step, scales
    positive, registry keyed by the master weight."""
    from runbooks_amd.ops.linear import (_FP8_REGISTRY, dequantize_fp8,
                                         quantize_fp8)

    torch.manual_seed(0)
    w = torch.randn(32, 64) * 3.0
    w8, scale = quantize_fp8(w)
    assert w8.dtype == torch.uint8 and scale.shape == (32,)
    assert (scale > 0).all()
    back = dequantize_fp8(w8, scale, dtype=torch.float32)
    # e4m3 has a 3-bit mantissa: relative error <= 2^-4 per element
    # against the row scale's dynamic range
    err = (back - w).abs()
    bound = w.abs().amax(dim=1, keepdim=True) / 448.0 + w.abs() * (2 ** -4)
    assert (err <= bound + 1e-6).all(), float((err - bound).max())
    assert w.data_ptr() in _FP8_REGISTRY
    _FP8_REGISTRY.pop(w.data_ptr())


def test_admission_does_not_starve_decode():
    """Merged-step scheduling: while new prompts are being admitted,
    already-running sequences still produce a token every step."""
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64,
                 max_batch=8)
    first = eng.submit([1, 2, 3], max_new_tokens=32)
    eng.step()  # prefill + first decode
    produced = len(first.output_ids)
    for i in range(4):  # four more admissions, one per step
        eng.submit([5 + i] * 4, max_new_tokens=8)
        eng.step()
        assert len(first.output_ids) > produced, "decode starved"
        produced = len(first.output_ids)


def test_prefix_cache_env_gate(monkeypatch):
    m = build_model("tiny-llama", dtype=torch.float32)
    # default ON since the r2 GPU validation pass
    assert Engine(m, device="cpu", dtype=torch.float32,
                  kv_blocks=32).prefix_cache_enabled
    monkeypatch.setenv("RB_PREFIX_CACHE", "0")
    assert not Engine(m, device="cpu", dtype=torch.float32,
                      kv_blocks=32).prefix_cache_enabled
    # explicit arg beats the env
    assert not Engine(m, device="cpu", dtype=torch.float32, kv_blocks=32,
                      prefix_cache=False).prefix_cache_enabled


def test_near_max_length_prompt_fits_block_table():
    """Regression: a prompt whose length lands in the
    last block window must not overflow max_blocks_per_seq — _admit
    allocates ceil(S/bs)+1 blocks, so the staging width needs the +1."""
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    bs, cap = eng.bs, eng.cfg.max_seq_len            # 16, 128
    S = cap - 2                                       # 126: last block window
    req = eng.submit(list(range(S % eng.cfg.vocab_size)) [:S] or [1],
                     max_new_tokens=2)
    req.prompt_ids = [i % eng.cfg.vocab_size for i in range(S)]
    while not req.finished:
        eng.step()
    assert len(req.output_ids) >= 1
    # block list may legitimately reach ceil(cap/bs)+1; staging must cover
    assert eng.max_blocks_per_seq >= (cap + bs - 1) // bs + 1


def test_prefill_failure_releases_blocks():
    """Regression: a prefill that raises must not leak
    the blocks _admit already allocated."""
    m = build_model("tiny-llama", dtype=torch.float32)
    eng = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64)
    free_before = len(eng.allocator.free)
    eng.submit([1, 2, 3], max_new_tokens=2)
    orig = eng._prefill
    def boom(req):
        raise RuntimeError("injected prefill failure")
    eng._prefill = boom
    with pytest.raises(RuntimeError, match="injected"):
        eng.step()
    assert len(eng.allocator.free) == free_before
    assert not eng.running and not eng.waiting
    # engine recovers: next request goes through untouched
    eng._prefill = orig
    r = eng.submit([4, 5, 6], max_new_tokens=2)
    while not r.finished:
        eng.step()
    assert len(r.output_ids) == 2


def test_multi_prefill_admission():
    """max_prefills_per_step > 1: a burst admits several prompts in one
    step (HTTP serving path) with identical outputs to serial admission."""
    m = build_model("tiny-llama", dtype=torch.float32)
    fast = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=5)
    fast.max_prefills_per_step = 4
    slow = Engine(m, device="cpu", dtype=torch.float32, kv_blocks=64, seed=5)
    prompts = [[1 + i, 9, 2] for i in range(6)]
    fr = [fast.submit(list(p), max_new_tokens=4) for p in prompts]
    sr = [slow.submit(list(p), max_new_tokens=4) for p in prompts]
    fast.step()
    assert len(fast.running) == 4      # burst admitted in one step
    while fast.has_work():
        fast.step()
    while slow.has_work():
        slow.step()
    for a, b in zip(fr, sr):
        assert a.output_ids == b.output_ids, (a.request_id, a.output_ids,
                                              b.output_ids)


@pytest.mark.parametrize("seed", [11])
def test_engine_fuzz_vt_gqa_bf16(seed):
    """The same prefix-cache + sliding-window fuzz invariants over a
    bf16 GQA model that allocates the TRANSPOSED-V cache on CPU (the
    reference kernels read it): allocator refcounts, free-list and
    window-drop bookkeeping are layout-independent."""
    import dataclasses
    import random

    from runbooks_amd.ops.attention import _is_vt

    cfg = dataclasses.replace(get_config("tiny-llama"), name="tiny-vt-gqa",
                              num_heads=8, num_kv_heads=2, head_dim=None,
                              hidden_size=512, sliding_window=32)
    m = build_model(cfg, dtype=torch.bfloat16)
    eng = Engine(m, device="cpu", dtype=torch.bfloat16, kv_blocks=28,
                 max_batch=4, seed=5, prefix_cache=True)
    assert _is_vt(*eng.caches[0][:2]), "GQA bf16 must allocate vt caches"
    rng = random.Random(seed)
    templates = [[i + 1] * 24 for i in range(2)]
    live = []
    for step in range(150):
        if rng.random() < 0.3 and len(live) < 12:
            p = list(rng.choice(templates)) + \
                [rng.randrange(256) for _ in range(rng.randint(0, 8))]
            live.append(eng.submit(p, max_new_tokens=rng.randint(1, 20),
                                   temperature=rng.choice([0.0, 0.8])))
        if eng.has_work():
            eng.step()
        counts = [0] * eng.allocator.num_blocks
        for r in eng.running:
            assert r.dropped % eng.bs == 0
            for b in r.blocks:
                counts[b] += 1
        for b in eng._pc.values():
            counts[b] += 1
        assert counts == eng.allocator.refs, (step, counts)
    for _ in range(2000):
        if not eng.has_work():
            break
        eng.step()
    assert not eng.has_work()
    eng.flush_prefix_cache()
    assert len(eng.allocator.free) == eng.allocator.num_blocks
