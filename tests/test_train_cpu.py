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
