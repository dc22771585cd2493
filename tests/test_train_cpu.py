"""Trainer loop + checkpoint/resume on CPU."""
import torch

from runbooks_amd.train import (TrainConfig, Trainer, latest_checkpoint,
                                load_checkpoint, save_checkpoint)


def test_loss_decreases(tmp_path):
    cfg = TrainConfig(model="tiny-llama", seq_len=32, micro_batch=4,
                      num_train_steps=15, dtype="float32", lr=5e-3,
                      output_dir=str(tmp_path))
    tr = Trainer(cfg)
    torch.manual_seed(0)
    batch = torch.randint(0, 256, (4, 33))
    losses = [tr.train_step(batch) for _ in range(15)]
    assert losses[-1] < losses[0] - 0.5, losses


def test_checkpoint_roundtrip_and_prune(tmp_path):
    state = {"a": torch.randn(4, 4), "b": torch.randn(8)}
    for step in (5, 10, 15, 20):
        save_checkpoint(tmp_path, step, state, keep=2)
    ckpts = sorted(p.name for p in tmp_path.glob("checkpoint-*"))
    assert ckpts == ["checkpoint-15", "checkpoint-20"]
    latest = latest_checkpoint(tmp_path)
    assert latest.name == "checkpoint-20"
    loaded, optim, step = load_checkpoint(latest)
    assert step == 20
    assert torch.equal(loaded["a"], state["a"])


def test_trainer_save_resume(tmp_path):
    cfg = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                      num_train_steps=3, dtype="float32",
                      output_dir=str(tmp_path))
    tr = Trainer(cfg)
    batch = torch.randint(0, 256, (2, 17))
    for _ in range(3):
        tr.train_step(batch)
    tr.save()
    tr2 = Trainer(cfg)
    assert tr2.resume()
    assert tr2.step_num == 3


def test_lora_fused_backward_matches_eager():
    """_LoRAFused (GEMM-only backward) vs plain autograd over the same math."""
    import torch
    from runbooks_amd.train.lora import _LoRAFused

    torch.manual_seed(3)
    T, IN, OUT, r, s = 9, 32, 24, 4, 2.0
    x = torch.randn(T, IN, requires_grad=True)
    w = torch.randn(OUT, IN)
    a = torch.randn(r, IN, requires_grad=True)
    b = torch.randn(OUT, r, requires_grad=True)
    dy = torch.randn(T, OUT)

    y = _LoRAFused.apply(x, w, a, b, s)
    y.backward(dy)

    x2 = x.detach().clone().requires_grad_(True)
    a2 = a.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    y2 = x2 @ w.t() + s * (x2 @ a2.t()) @ b2.t()
    y2.backward(dy)

    assert torch.allclose(y, y2, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(a.grad, a2.grad, atol=1e-5)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)


def test_checkpoint_resume_with_optimizer(tmp_path):
    """save -> fresh Trainer -> resume restores step, weights, AND the
    AdamW moments (training continues instead of restarting)."""
    import torch
    from runbooks_amd.train import TrainConfig, Trainer

    cfg = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                      num_train_steps=3, save_steps=0, dtype="float32",
                      output_dir=str(tmp_path), seed=1)
    t1 = Trainer(cfg)
    batch = torch.randint(0, 256, (2, 17))
    for _ in range(3):
        t1.train_step(batch)
    t1.save()

    t2 = Trainer(cfg)
    assert t2.resume()
    assert t2.step_num == 3
    # lora weights restored
    sd1 = {k: v for k, v in t1.model.state_dict().items() if "lora" in k}
    sd2 = {k: v for k, v in t2.model.state_dict().items() if "lora" in k}
    for k in sd1:
        assert torch.equal(sd1[k], sd2[k]), k
    # optimizer moments restored (non-zero after 3 steps)
    states = list(t2.optimizer.state.values())
    assert states and any(s["exp_avg"].abs().sum() > 0 for s in states)
    assert all(s["step"] == 3 for s in states)
    # training continues
    t2.train_step(batch)
    assert t2.step_num == 4


def test_gradient_accumulation_matches_big_batch():
    """2 micro-batches with grad_accum_steps=2 == one batch of 2x size."""
    import torch
    from runbooks_amd.train import TrainConfig, Trainer

    torch.manual_seed(0)
    b1 = torch.randint(0, 256, (2, 17))
    b2 = torch.randint(0, 256, (2, 17))
    big = torch.cat([b1, b2])

    cfg_a = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                        num_train_steps=1, grad_accum_steps=2,
                        dtype="float32", grad_clip=0, seed=2)
    ta = Trainer(cfg_a)
    ta.train_step(b1, sync=False)
    ta.train_step(b2, sync=True)
    assert ta.step_num == 1

    cfg_b = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=4,
                        num_train_steps=1, grad_accum_steps=1,
                        dtype="float32", grad_clip=0, seed=2)
    tb = Trainer(cfg_b)
    tb.train_step(big)

    sa = {k: v for k, v in ta.model.state_dict().items() if "lora" in k}
    sb = {k: v for k, v in tb.model.state_dict().items() if "lora" in k}
    for k in sa:
        assert torch.allclose(sa[k], sb[k], atol=1e-5), k


def test_lr_schedule_warmup_cosine():
    import pytest
    import torch
    from runbooks_amd.train import TrainConfig, Trainer

    cfg = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                      num_train_steps=10, lr=1e-2, lr_scheduler="cosine",
                      warmup_steps=2, dtype="float32", seed=0)
    t = Trainer(cfg)
    b = torch.randint(0, 256, (2, 17))
    lrs = []
    for _ in range(10):
        t.train_step(b)
        lrs.append(t.optimizer.param_groups[0]["lr"])
    assert lrs[0] == pytest.approx(1e-2 * 0.5)   # warmup step 1/2
    assert lrs[1] == pytest.approx(1e-2)         # warmup done
    assert lrs[2] < lrs[1]                       # cosine decays
    assert lrs[-1] == pytest.approx(0.0, abs=1e-6)


def test_full_finetune_step():
    """full_finetune=True trains every parameter (no LoRA wrap)."""
    import torch
    from runbooks_amd.train import TrainConfig, Trainer

    cfg = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                      num_train_steps=1, full_finetune=True,
                      dtype="float32", seed=0)
    t = Trainer(cfg)
    n_trainable = sum(p.numel() for p in t.model.parameters()
                      if p.requires_grad)
    n_total = sum(p.numel() for p in t.model.parameters())
    assert n_trainable == n_total
    before = t.model.embed.weight.detach().clone()
    t.train_step(torch.randint(0, 256, (2, 17)))
    assert not torch.equal(before, t.model.embed.weight.detach())


def test_text_dataset_formats(tmp_path):
    """TextDataset reads jsonl / parquet / csv / txt (the formats
    dataset-loader jobs produce) and concatenates a directory."""
    import json as _json
    from runbooks_amd.train import TextDataset

    (tmp_path / "a.jsonl").write_text(
        _json.dumps({"text": "alpha"}) + "\n" +
        _json.dumps({"prompt": "b", "completion": "eta"}) + "\n")
    (tmp_path / "b.txt").write_text("gamma\ndelta\n")
    (tmp_path / "c.csv").write_text("text\nepsilon\n")
    import pyarrow as pa
    import pyarrow.parquet as pq
    pq.write_table(pa.table({"text": ["zeta", "eta"]}), tmp_path / "d.parquet")

    ds = TextDataset(tmp_path, seq_len=16, vocab_size=256)
    assert len(ds) == 7
    row = ds[0]
    assert row.shape == (16,) and row.dtype == torch.long
    single = TextDataset(tmp_path / "d.parquet", seq_len=8, vocab_size=256)
    assert len(single) == 2


def test_grad_checkpointing_matches_plain():
    """Activation checkpointing recomputes forward in backward: loss and
    gradients must be bitwise-equal to the plain path (fp32 CPU)."""
    from runbooks_amd.models import build_model

    tokens = torch.randint(0, 256, (2, 17))

    def run(ckpt):
        torch.manual_seed(0)
        m = build_model("tiny-llama", dtype=torch.float32, seed=7)
        if ckpt:
            m.enable_grad_checkpointing()
        logits = m(tokens[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), tokens[:, 1:].reshape(-1))
        loss.backward()
        grads = {n: p.grad.clone() for n, p in m.named_parameters()
                 if p.grad is not None}
        return float(loss.detach()), grads

    l0, g0 = run(False)
    l1, g1 = run(True)
    assert l0 == l1
    assert g0.keys() == g1.keys() and len(g0) > 0
    for n in g0:
        assert torch.equal(g0[n], g1[n]), n


def test_trainer_evaluate_and_eval_steps(capsys):
    """evaluate() returns a finite mean CE; fit() logs eval_loss at the
    configured cadence."""
    from runbooks_amd.train import SyntheticTokens, TrainConfig, Trainer

    cfg = TrainConfig(model="tiny-llama", seq_len=16, micro_batch=2,
                      num_train_steps=4, eval_steps=2, dtype="float32",
                      save_steps=0, output_dir="/tmp/rb-eval-test")
    tr = Trainer(cfg, device="cpu")
    ev = SyntheticTokens(tr.model.cfg.vocab_size, 17, n=8, seed=9)
    loss = tr.evaluate(ev, max_batches=2)
    assert loss == loss and loss > 0  # finite, positive
    tr.fit(SyntheticTokens(tr.model.cfg.vocab_size, 17, n=16), ev,
           log_every=100)
    out = capsys.readouterr().out
    assert out.count("eval_loss") == 2, out


def test_lora_dropout():
    """dropout>0: stochastic in train mode, exact (no-dropout) in eval;
    dropout=0 keeps the fused GEMM-only path."""
    from runbooks_amd.models import build_model
    from runbooks_amd.train import apply_lora

    torch.manual_seed(0)
    m = build_model("tiny-llama", dtype=torch.float32, seed=2)
    apply_lora(m, r=4, alpha=8, dropout=0.5)
    # make adapters non-zero so dropout actually matters
    for n, p in m.named_parameters():
        if "lora_b" in n:
            with torch.no_grad():
                p.add_(torch.randn_like(p) * 0.1)
    x = torch.randint(0, 256, (1, 8))
    m.train()
    a = m(x)
    b = m(x)
    assert not torch.allclose(a, b), "dropout inactive in train mode"
    m.eval()
    with torch.no_grad():
        c, d = m(x), m(x)
    assert torch.equal(c, d)


def test_lora_group_matches_ungrouped(monkeypatch):
    """Grouped q/k/v + gate/up A-GEMMs (one activation stream per group)
    produce the SAME loss and lora grads as the per-module fused path."""
    import torch

    from runbooks_amd.models import build_model
    from runbooks_amd.train.lora import apply_lora

    def run(grouped: bool):
        monkeypatch.setenv("RB_LORA_GROUP", "1" if grouped else "0")
        torch.manual_seed(11)
        m = build_model("tiny-llama", dtype=torch.float32, seed=5)
        apply_lora(m, r=4, alpha=8)
        has_group = any(getattr(mm, "_group", None) is not None
                        for mm in m.modules())
        assert has_group == grouped
        torch.manual_seed(2)
        x = torch.randint(0, m.cfg.vocab_size, (2, 12))
        logits = m(x)
        loss = logits.float().pow(2).mean()
        loss.backward()
        grads = {k: p.grad.clone() for k, p in m.named_parameters()
                 if p.grad is not None}
        return float(loss), grads

    l1, g1 = run(True)
    l2, g2 = run(False)
    assert abs(l1 - l2) < 1e-6, (l1, l2)
    assert set(g1) == set(g2) and g1, "lora grads must exist"
    for k in g1:
        assert torch.allclose(g1[k], g2[k], atol=1e-5), (k, (g1[k] - g2[k]).abs().max())


def test_lora_group_cache_does_not_leak_graph():
    """The group tcat cache must be dropped after every member consumed
    it (a retained autograd graph across steps is a memory leak)."""
    import torch

    from runbooks_amd.models import build_model
    from runbooks_amd.train.lora import apply_lora, _LoRAGroup

    torch.manual_seed(0)
    m = build_model("tiny-llama", dtype=torch.float32, seed=5)
    apply_lora(m, r=4, alpha=8)
    x = torch.randint(0, m.cfg.vocab_size, (2, 8))
    m(x).sum().backward()
    groups = {id(mm._group): mm._group for mm in m.modules()
              if getattr(mm, "_group", None) is not None}
    assert groups
    for g in groups.values():
        assert isinstance(g, _LoRAGroup)
        assert g._t is None and g._left == 0


def test_grad_checkpointing_with_lora_groups():
    """Activation checkpointing recomputes the block forward with a
    FRESH activation tensor each time — the LoRA group tcat cache
    (keyed on tensor identity) must not serve stale tensors across the
    recompute; grads must match the non-checkpointed run exactly."""
    import torch

    from runbooks_amd.models import build_model
    from runbooks_amd.train.lora import apply_lora

    tokens = torch.randint(0, 256, (2, 17))

    def run(ckpt):
        torch.manual_seed(0)
        m = build_model("tiny-llama", dtype=torch.float32, seed=7)
        apply_lora(m, r=4, alpha=8)
        assert any(getattr(mm, "_group", None) is not None
                   for mm in m.modules())
        if ckpt:
            m.enable_grad_checkpointing()
        logits = m(tokens[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), tokens[:, 1:].reshape(-1))
        loss.backward()
        grads = {n: p.grad.clone() for n, p in m.named_parameters()
                 if p.grad is not None}
        return float(loss.detach()), grads

    l0, g0 = run(False)
    l1, g1 = run(True)
    assert l0 == l1
    assert g0.keys() == g1.keys() and len(g0) > 0
    for n in g0:
        assert torch.equal(g0[n], g1[n]), n
