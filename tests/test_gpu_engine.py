"""GPU engine-level tests: hipGraph decode correctness and the serving
fast paths (fused qkv epilogue, packed swiglu) against the eager path."""
import pytest
import torch

from runbooks_amd import ops
from runbooks_amd.models import build_model
from runbooks_amd.serve import Engine

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_graph_decode_matches_eager():
    assert ops.has_hip()
    torch.manual_seed(0)
    prompt = [5, 3, 8, 1, 9, 2]

    eng_g = Engine(build_model("smoke-llama", dtype=torch.bfloat16, seed=4),
                   device=DEV, kv_blocks=128, seed=11)
    assert eng_g.use_graphs
    out_g = eng_g.generate(list(prompt), max_new_tokens=8)

    eng_e = Engine(build_model("smoke-llama", dtype=torch.bfloat16, seed=4),
                   device=DEV, kv_blocks=128, seed=11)
    eng_e.use_graphs = False
    out_e = eng_e.generate(list(prompt), max_new_tokens=8)
    assert out_g == out_e, (out_g, out_e)

    # replay determinism: same prompt again through the cached graph
    out_g2 = eng_g.generate(list(prompt), max_new_tokens=8)
    assert out_g2 == out_g


def test_continuous_batching_gpu():
    assert ops.has_hip()
    eng = Engine(build_model("smoke-llama", dtype=torch.bfloat16, seed=4),
                 device=DEV, kv_blocks=256, seed=13)
    reqs = [eng.submit([1 + i, 2, 3], max_new_tokens=5 + i) for i in range(4)]
    while eng.has_work():
        eng.step()
    for i, r in enumerate(reqs):
        assert r.finished and len(r.output_ids) == 5 + i
    # all KV blocks returned to the allocator
    assert len(eng.allocator.free) == eng.allocator.num_blocks


def test_fp8_engine_decode_runs():
    """MODEL_LOAD_IN_8BIT path: fp8 decode generates plausibly
    (weight-only e4m3 shifts logits, so compare shape/validity, plus the
    fp8 registry actually being used)."""
    assert ops.has_hip()
    from runbooks_amd.ops.linear import _FP8_REGISTRY
    _FP8_REGISTRY.clear()
    # smoke-llama dims satisfy the fp8 kernel constraints (K%256, N%64)
    m = build_model("smoke-llama", dtype=torch.bfloat16, seed=1)
    eng = Engine(m, device=DEV, kv_blocks=128, seed=3, load_in_8bit=True)
    assert len(_FP8_REGISTRY) > 0, "fp8 weights not registered"
    out = eng.generate([10, 20, 30], max_new_tokens=4)
    assert len(out) == 4 and all(0 <= t < m.cfg.vocab_size for t in out)
    _FP8_REGISTRY.clear()
