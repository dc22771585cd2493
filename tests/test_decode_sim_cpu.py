"""CPU simulation of ops/csrc/attention_decode.hip's dataflow.

Mirrors the kernel's exact structure — the (wave, 16-lane-group) token
striding, per-stream online softmax, the shfl group merge, the LDS wave
merge, the split-seq partial (m, l, acc) slabs and the combine kernel —
in numpy, and checks it against ops.paged_decode_ref. The kernel is
GPU-validated (tests/test_gpu_ops.py::test_paged_decode); this sim
documents its index math so changes (e.g. the round-2 per-sequence
``seq_start`` for strict sliding windows) can be logic-checked before
spending GPU budget.

Keep in sync with attention_decode.hip when editing either.
"""
import math

import numpy as np
import pytest
import torch

NW, GRPS = 4, 4  # 4 waves x 4 sixteen-lane groups per workgroup


def _merge(a, b):
    """Flash-style merge of two (m, l, acc) states."""
    m1, l1, a1 = a
    m2, l2, a2 = b
    mn = max(m1, m2)
    if mn == -math.inf:
        return a
    e1, e2 = math.exp(m1 - mn), math.exp(m2 - mn)
    return mn, l1 * e1 + l2 * e2, a1 * e1 + a2 * e2


def simulate_decode(q, kc, vc, bt, seq_lens, scale, nsplit=1,
                    seq_starts=None):
    """q [B, Hq, Dh]; kc/vc [nblk, Hkv, BS, Dh]; bt [B, max_blocks].
    float64 numpy throughout (the kernel's fp32 accum, idealized).

    ``seq_starts`` models the planned round-2 kernel extension for
    STRICT sliding windows: attend cache positions [start, seq) only.
    Kernel-side this is two lines — ``len = seq_len - start; chunk =
    ceil(len/nsplit); t_begin = start + split*chunk`` — validated here
    against a masked reference before any GPU budget is spent."""
    B, Hq, Dh = q.shape
    hkv, bs = kc.shape[1], kc.shape[2]
    G = Hq // hkv
    out = np.zeros_like(q)
    for b in range(B):
        seq = int(seq_lens[b])
        start = int(seq_starts[b]) if seq_starts is not None else 0
        for h in range(hkv):
            for g in range(G):
                qv = q[b, h * G + g] * scale
                parts = []
                for split in range(nsplit):
                    chunk = (seq - start + nsplit - 1) // nsplit
                    t0 = start + split * chunk
                    t1 = min(seq, t0 + chunk)
                    # one online-softmax stream per (wave, group), tokens
                    # t0 + wid*4 + grp, stride 16 — the kernel's layout
                    streams = []
                    for wid in range(NW):
                        for grp in range(GRPS):
                            m, l = -math.inf, 0.0
                            acc = np.zeros(Dh)
                            for t in range(t0 + wid * 4 + grp, t1, NW * 4):
                                blk = bt[b][t // bs]
                                k = kc[blk, h, t % bs]
                                v = vc[blk, h, t % bs]
                                s = float(qv @ k)
                                mn = max(m, s)
                                alpha = math.exp(m - mn) \
                                    if m != -math.inf else 0.0
                                p = math.exp(s - mn)
                                l = l * alpha + p
                                acc = acc * alpha + p * v
                                m = mn
                            streams.append((m, l, acc))
                    # group merge (shfl_xor 16/32) then wave merge (LDS):
                    # order-independent flash merges
                    state = streams[0]
                    for s in streams[1:]:
                        state = _merge(state, s)
                    parts.append(state)
                # decode_combine_kernel: merge the split partials
                state = parts[0]
                for s in parts[1:]:
                    state = _merge(state, s)
                m, l, acc = state
                out[b, h * G + g] = acc / l if l > 0 else 0.0
    return out


def _case(B, hkv, G, bs, max_blocks, seqs, dh, seed):
    rng = np.random.default_rng(seed)
    Hq = hkv * G
    q = rng.standard_normal((B, Hq, dh))
    nblk = B * max_blocks + 1
    kc = rng.standard_normal((nblk, hkv, bs, dh))
    vc = rng.standard_normal((nblk, hkv, bs, dh))
    # distinct blocks per row, deliberately non-monotone (window trims /
    # prefix sharing produce arbitrary tables)
    ids = rng.permutation(nblk - 1)[: B * max_blocks].reshape(B, max_blocks)
    return q, kc, vc, ids, np.array(seqs), 1.0 / math.sqrt(dh)


@pytest.mark.parametrize("B,hkv,G,seqs,nsplit", [
    (2, 2, 1, [5, 33], 1),          # MHA, ragged
    (1, 1, 8, [40], 1),             # MQA G=8 (falcon-ish)
    (2, 2, 4, [64, 17], 4),         # GQA + split-seq partials
    (1, 2, 2, [3], 8),              # more splits than tokens: empty splits
])
def test_decode_sim_matches_ref(B, hkv, G, seqs, nsplit):
    from runbooks_amd.ops.attention import paged_decode_ref

    bs, dh = 16, 64
    maxb = (max(seqs) + bs - 1) // bs
    q, kc, vc, bt, sl, scale = _case(B, hkv, G, bs, maxb, seqs, dh, B + G)
    got = simulate_decode(q, kc, vc, bt, sl, scale, nsplit=nsplit)

    ref = paged_decode_ref(
        torch.tensor(q, dtype=torch.float32),
        torch.tensor(kc, dtype=torch.float32),
        torch.tensor(vc, dtype=torch.float32),
        torch.tensor(bt, dtype=torch.int32),
        torch.tensor(sl, dtype=torch.int32), scale).numpy()
    assert np.allclose(got, ref, atol=1e-5), np.abs(got - ref).max()


def test_decode_sim_seq_start_strict_window():
    """The seq_start extension (round-2 strict sliding window): the sim
    attends only [start, seq) and matches a sliced reference — including
    a start mid-block and start==0 rows mixed in one batch."""
    from runbooks_amd.ops.attention import paged_decode_ref

    bs, dh = 16, 64
    q, kc, vc, bt, sl, scale = _case(3, 2, 2, bs, 4, [50, 61, 9], dh, 7)
    starts = np.array([18, 0, 5])  # mid-block, none, mid-block
    got = simulate_decode(q, kc, vc, bt, sl, scale, nsplit=4,
                          seq_starts=starts)
    # reference: shift each row's window to the front of a fresh cache
    for b in range(3):
        n = int(sl[b]) - int(starts[b])
        kc2 = np.zeros((4, 2, bs, dh))
        vc2 = np.zeros((4, 2, bs, dh))
        for i in range(n):
            t = int(starts[b]) + i
            blk = bt[b][t // bs]
            kc2[i // bs, :, i % bs] = kc[blk, :, t % bs]
            vc2[i // bs, :, i % bs] = vc[blk, :, t % bs]
        ref = paged_decode_ref(
            torch.tensor(q[b:b + 1], dtype=torch.float32),
            torch.tensor(kc2, dtype=torch.float32),
            torch.tensor(vc2, dtype=torch.float32),
            torch.tensor([[0, 1, 2, 3]], dtype=torch.int32),
            torch.tensor([n], dtype=torch.int32), scale).numpy()
        assert np.allclose(got[b], ref[0], atol=1e-5), b


def test_engine_cpu_bf16_gqa_transposed_v_cache():
    """A bf16 GQA (G=4) model on CPU allocates the transposed-V cache
    and the reference decode path reads it: engine greedy decode must
    match the full no-cache forward (end-to-end vt coverage without a
    GPU)."""
    import dataclasses

    import torch

    from runbooks_amd.models import build_model, get_config
    from runbooks_amd.models.config import register
    from runbooks_amd.ops.attention import _is_vt
    from runbooks_amd.serve import Engine

    cfg = dataclasses.replace(get_config("smoke-llama"), num_heads=8,
                              num_kv_heads=2, head_dim=None, hidden_size=512,
                              name="smoke-llama-gqa4")
    register(cfg)
    m = build_model(cfg.name, dtype=torch.bfloat16, seed=9)
    eng = Engine(m, device="cpu", kv_blocks=64, seed=1)
    assert _is_vt(*eng.caches[0][:2]), "GQA bf16 must allocate vt caches"
    prompt = [3, 1, 4, 1, 5]
    out = eng.generate(list(prompt), max_new_tokens=5)
    seq = list(prompt)
    for _ in range(5):
        with torch.no_grad():
            logits = m(torch.tensor([seq]))
        seq.append(int(logits[0, -1].float().argmax()))
    assert out == seq[len(prompt):], (out, seq[len(prompt):])
