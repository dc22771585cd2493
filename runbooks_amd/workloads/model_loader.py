"""model-loader image main.

Parity: substratusai/model-loader-huggingface (reference
examples/facebook-opt-125m/base-model.yaml:7-9) — downloads the HF model
named by PARAM_NAME into /content/artifacts as safetensors + config +
tokenizer files, which the trainer/server then mount at /content/model.

Offline mode (PARAM_SYNTHETIC=true or no network): materializes a
random-init checkpoint of a registry architecture instead, so the whole
platform runs air-gapped (bench/system-test path).
"""
from __future__ import annotations

import json
import os
import sys
from pathlib import Path


def load_from_hub(name: str, out_dir: Path) -> None:
    from huggingface_hub import snapshot_download
    snapshot_download(
        repo_id=name, local_dir=str(out_dir),
        allow_patterns=["*.safetensors", "*.json", "tokenizer.model",
                        "*.txt"],
        token=os.environ.get("HUGGING_FACE_HUB_TOKEN") or None)


def materialize_synthetic(name: str, out_dir: Path, seed: int = 0) -> None:
    import torch
    from safetensors.torch import save_file

    from ..models import build_model, get_config

    cfg = get_config(name)
    model = build_model(cfg, dtype=torch.bfloat16, tp=1, seed=seed)
    save_file({k: v.contiguous() for k, v in model.state_dict().items()
               if not k.startswith("rope_")},
              str(out_dir / "model.safetensors"))
    (out_dir / "config.json").write_text(json.dumps({
        "runbooks_amd_config": cfg.name}))


def main():
    name = os.environ.get("PARAM_NAME", "")
    if not name:
        print("model-loader: PARAM_NAME is required", file=sys.stderr)
        return 2
    out_dir = Path(os.environ.get("ARTIFACTS_DIR", "/content/artifacts"))
    out_dir.mkdir(parents=True, exist_ok=True)
    synthetic = os.environ.get("PARAM_SYNTHETIC", "").lower() in ("1", "true")
    if synthetic:
        materialize_synthetic(name.split("/")[-1], out_dir)
    else:
        try:
            load_from_hub(name, out_dir)
        except Exception as e:  # air-gapped fallback
            print(f"model-loader: hub download failed ({e}); "
                  f"trying synthetic registry model", file=sys.stderr)
            materialize_synthetic(name.split("/")[-1], out_dir)
    (out_dir / "completed.json").write_text(
        json.dumps({"completed": True}))
    print(f"model-loader: wrote {sorted(p.name for p in out_dir.iterdir())}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
