"""server image main.

Parity: substratusai/model-server-basaran (reference
examples/llama2-7b/server.yaml, container-contract.md:50-56) — loads
/content/model, serves an OpenAI-style completions API on :8080 with
"/" readiness. Honors the contract env knobs:

- MODEL_LOAD_IN_8BIT / MODEL_LOAD_IN_4BIT: weight quantization hints
  (reference examples/llama2-7b/server.yaml:9). This runtime maps 8-bit
  to fp8 storage when available and otherwise serves bf16 (288 GB HBM3E
  rarely needs it).
- TP: tensor parallelism degree (defaults to visible GPU count); ranks
  are spawned via torch.distributed.run over RCCL/xGMI.
"""
from __future__ import annotations

import json
import os
import sys
from pathlib import Path


def _maybe_relaunch_tp() -> bool:
    import torch
    n = int(os.environ.get("TP", "0")) or (
        torch.cuda.device_count() if torch.cuda.is_available() else 1)
    if n <= 1 or os.environ.get("RANK") is not None:
        return False
    os.execvp(sys.executable, [
        sys.executable, "-m", "torch.distributed.run",
        "--standalone", "--local-addr", "127.0.0.1",
        f"--nproc-per-node={n}",
        "-m", "runbooks_amd.workloads.server_main"])
    return True


def main():
    if _maybe_relaunch_tp():
        return 0
    import torch

    from ..models import build_model, get_config, list_configs
    from ..models.load import config_from_hf_json, load_pretrained
    from ..parallel import comm
    from ..serve import Engine
    from ..serve.http import serve_forever
    from ..serve.tokenizer import load_tokenizer

    comm.init_from_env()
    model_dir = Path(os.environ.get("MODEL_DIR", "/content/model"))
    # A fine-tuned Model's artifacts hold trainer checkpoints
    # (checkpoint-N/model.safetensors + meta.json) rather than top-level
    # files: serve the latest checkpoint (the Server CRD points at the
    # Model's artifacts — reference server_controller.go model mount).
    if not list(model_dir.glob("*.safetensors")):
        ckpts = sorted(model_dir.glob("checkpoint-*"),
                       key=lambda p: int(p.name.split("-")[-1]))
        if ckpts:
            model_dir = ckpts[-1]
            print(f"server: serving latest checkpoint {model_dir}")
    arch = os.environ.get("PARAM_MODEL") or "llama2-7b"
    marker = model_dir / "config.json"
    cfg = None
    if marker.exists():
        meta = json.loads(marker.read_text())
        name = meta.get("runbooks_amd_config")
        if name and name in list_configs():
            cfg = get_config(name)
        elif "runbooks_amd_fields" in meta:
            from ..models import ModelConfig
            from ..models.config import register
            cfg = register(ModelConfig(**meta["runbooks_amd_fields"]))
        elif "model_type" in meta or "architectures" in meta:
            cfg = config_from_hf_json(marker)
    if cfg is None:
        cfg = get_config(arch)

    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    model = build_model(cfg, dtype=dtype)
    if model_dir.exists() and list(model_dir.glob("*.safetensors")):
        load_pretrained(model, model_dir, rank=comm.rank(),
                        tp=comm.world_size(), strict=False)
        print(f"server: loaded weights from {model_dir}")

    in_8bit = os.environ.get("MODEL_LOAD_IN_8BIT", "").lower() in ("1", "true")
    # PARAM_KV_FP8 / KV_FP8: e4m3 KV cache (2x capacity, long-context)
    kv_fp8 = os.environ.get("PARAM_KV_FP8",
                            os.environ.get("KV_FP8", "")).lower() in \
        ("1", "true")
    engine = Engine(model, load_in_8bit=in_8bit, kv_fp8=kv_fp8)
    tok = load_tokenizer(model_dir if model_dir.exists() else None)
    if comm.rank() == 0:
        serve_forever(engine, tok, port=int(os.environ.get("PORT", "8080")),
                      model_name=cfg.name)
    else:
        # TP worker ranks follow rank 0's collectives inside the model;
        # they loop in the engine's worker protocol.
        from ..serve.tp_worker import worker_loop
        worker_loop(engine)
    return 0


if __name__ == "__main__":
    sys.exit(main())
