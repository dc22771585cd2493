"""Checkpoint / resume to the artifact directory.

The reference delegates checkpointing to the HF trainer image writing
into the bucket-mounted /content/artifacts (save_steps param, reference
examples/llama2-7b/finetuned-model.yaml:14-16; design rationale in the
reference docs/design.md "Buckets"). Here it is native: safetensors
weights + a json manifest, atomic rename, resume-from-latest.
"""
from __future__ import annotations

import json
import os
import tempfile
from pathlib import Path

import torch
from safetensors.torch import load_file, save_file


def save_checkpoint(out_dir: str | Path, step: int, model_state: dict,
                    optim_state: dict | None = None, keep: int = 3,
                    model_cfg=None) -> Path:
    out_dir = Path(out_dir)
    out_dir.mkdir(parents=True, exist_ok=True)
    ckpt = out_dir / f"checkpoint-{step}"
    tmp = Path(tempfile.mkdtemp(dir=out_dir, prefix=".tmp-ckpt-"))
    save_file({k: v.contiguous().cpu() for k, v in model_state.items()},
              str(tmp / "model.safetensors"))
    if optim_state is not None:
        torch.save(optim_state, tmp / "optimizer.pt")
    (tmp / "meta.json").write_text(json.dumps({"step": step}))
    if model_cfg is not None:
        # the serving contract: a Server pointed at these artifacts must
        # be able to reconstruct the architecture (reference flow:
        # finetuned Model -> Server mounts its artifacts)
        import dataclasses
        (tmp / "config.json").write_text(json.dumps({
            "runbooks_amd_config": model_cfg.name,
            "runbooks_amd_fields": dataclasses.asdict(model_cfg)}))
    if ckpt.exists():
        import shutil
        shutil.rmtree(ckpt)
    os.rename(tmp, ckpt)
    _prune(out_dir, keep)
    return ckpt


def _prune(out_dir: Path, keep: int):
    ckpts = sorted(out_dir.glob("checkpoint-*"),
                   key=lambda p: int(p.name.split("-")[-1]))
    for p in ckpts[:-keep]:
        import shutil
        shutil.rmtree(p, ignore_errors=True)


def latest_checkpoint(out_dir: str | Path) -> Path | None:
    out_dir = Path(out_dir)
    if not out_dir.exists():
        return None
    ckpts = sorted(out_dir.glob("checkpoint-*"),
                   key=lambda p: int(p.name.split("-")[-1]))
    return ckpts[-1] if ckpts else None


def load_checkpoint(ckpt: str | Path, device="cpu"):
    ckpt = Path(ckpt)
    state = load_file(str(ckpt / "model.safetensors"), device=device)
    meta = json.loads((ckpt / "meta.json").read_text())
    optim_path = ckpt / "optimizer.pt"
    optim = torch.load(optim_path, map_location=device, weights_only=False) \
        if optim_path.exists() else None
    return state, optim, meta["step"]
