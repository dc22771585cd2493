"""trainer image main.

Parity: substratusai/model-trainer-huggingface (reference
examples/llama2-7b/finetuned-model.yaml:11-16) — fine-tunes the base
model mounted at /content/model on the dataset at /content/data, writing
checkpoints to /content/artifacts. Training knobs arrive as PARAM_* env
(the contract's TrainingArguments passthrough): num_train_epochs /
num_train_steps, save_steps, learning_rate, per_device_train_batch_size,
seq_len, lora_r, full_finetune.

Multi-GPU: DP over RCCL/xGMI — when GPU_COUNT/amd.com/gpu > 1 the main
re-execs itself under torch.distributed.run with one rank per GPU.
"""
from __future__ import annotations

import json
import os
import sys
from pathlib import Path


def _param(name: str, default=None, cast=str):
    v = os.environ.get(f"PARAM_{name.upper()}")
    if v is None:
        return default
    if cast is bool:
        return v.lower() in ("1", "true", "yes")
    return cast(v)


def _gpu_count() -> int:
    import torch
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def _maybe_relaunch_distributed() -> bool:
    """Re-exec under torchrun for DP when >1 GPU is visible."""
    n = _gpu_count()
    if n <= 1 or os.environ.get("RANK") is not None:
        return False
    os.execvp(sys.executable, [
        sys.executable, "-m", "torch.distributed.run",
        "--standalone", "--local-addr", "127.0.0.1",
        f"--nproc-per-node={n}",
        "-m", "runbooks_amd.workloads.trainer_main"])
    return True  # unreachable


def main():
    if _maybe_relaunch_distributed():
        return 0
    import torch

    from ..models import list_configs
    from ..models.load import load_pretrained
    from ..parallel import comm
    from ..serve.tokenizer import load_tokenizer
    from ..train import TrainConfig, Trainer
    from ..train.data import TEXT_SUFFIXES, SyntheticTokens, TextDataset

    comm.init_from_env()
    model_dir = Path(os.environ.get("MODEL_DIR", "/content/model"))
    data_dir = Path(os.environ.get("DATA_DIR", "/content/data"))
    out_dir = os.environ.get("ARTIFACTS_DIR", "/content/artifacts")

    # architecture: explicit param, the loader's marker, or llama2-7b
    arch = _param("model") or "llama2-7b"
    marker = model_dir / "config.json"
    if marker.exists():
        meta = json.loads(marker.read_text())
        arch = meta.get("runbooks_amd_config", arch)
        if arch not in list_configs():
            from ..models.load import config_from_hf_json
            from ..models import register
            arch = config_from_hf_json(marker)
            # Trainer resolves models by name via the registry; a config
            # derived from an out-of-registry HF checkpoint must be
            # registered or build_model(name) raises KeyError at startup
            # (server_main passes the config object through instead).
            register(arch)

    cfg = TrainConfig(
        model=arch if isinstance(arch, str) else arch.name,
        seq_len=_param("seq_len", 512, int),
        micro_batch=_param("per_device_train_batch_size", 4, int),
        lr=_param("learning_rate", 2e-4, float),
        lr_scheduler=_param("lr_scheduler_type", "constant"),
        warmup_steps=_param("warmup_steps", 0, int),
        num_train_steps=_param("num_train_steps", 100, int),
        save_steps=_param("save_steps", 50, int),
        grad_accum_steps=_param("gradient_accumulation_steps", 1, int),
        lora_r=_param("lora_r", 16, int),
        lora_dropout=_param("lora_dropout", 0.0, float),
        full_finetune=_param("full_finetune", False, bool),
        grad_checkpointing=_param("gradient_checkpointing", False, bool),
        eval_steps=_param("eval_steps", 0, int),
        dtype="bfloat16" if torch.cuda.is_available() else "float32",
        output_dir=out_dir)
    trainer = Trainer(cfg)

    if (model_dir / "model.safetensors").exists() or \
            list(model_dir.glob("*.safetensors")):
        load_pretrained(trainer.model, model_dir, rank=0, tp=1, strict=False)
        print(f"trainer: loaded base weights from {model_dir}")

    files = (sorted(p for p in data_dir.iterdir()
                    if p.suffix.lower() in TEXT_SUFFIXES)
             if data_dir.exists() else [])
    if files:
        tok = load_tokenizer(model_dir if model_dir.exists() else None)
        dataset = TextDataset(data_dir, cfg.seq_len + 1, tokenizer=tok,
                              vocab_size=trainer.model.cfg.vocab_size)
        epochs = _param("num_train_epochs", None, float)
        if epochs is not None:
            steps = int(epochs * len(dataset)
                        / (cfg.micro_batch * comm.world_size()))
            trainer.cfg.num_train_steps = max(1, steps)
    else:
        dataset = SyntheticTokens(trainer.model.cfg.vocab_size,
                                  cfg.seq_len + 1)

    eval_dataset = None
    if trainer.cfg.eval_steps and files:
        # hold out ~5% of rows (HF-style eval split) for eval_loss
        n_eval = max(1, len(dataset) // 20)
        eval_dataset = torch.utils.data.Subset(
            dataset, range(len(dataset) - n_eval, len(dataset)))
        dataset = torch.utils.data.Subset(
            dataset, range(len(dataset) - n_eval))

    trainer.resume()
    trainer.fit(dataset, eval_dataset=eval_dataset)
    if comm.rank() == 0:
        # Final SERVABLE model at the artifacts root: merge LoRA into
        # the base and save full weights + the architecture marker —
        # a Server pointed at this Model's artifacts loads it directly
        # (reference flow: finetuned-model.yaml -> server.yaml mounts
        # the model at /content/model). Intermediate checkpoint-N dirs
        # stay adapter-only for cheap resume.
        import dataclasses

        from safetensors.torch import save_file

        from ..train.lora import merge_lora
        if not cfg.full_finetune:
            merge_lora(trainer.model)
        save_file({k: v.contiguous().cpu()
                   for k, v in trainer.model.state_dict().items()
                   if not k.startswith("rope_")},
                  str(Path(out_dir) / "model.safetensors"))
        (Path(out_dir) / "config.json").write_text(json.dumps({
            "runbooks_amd_config": trainer.model.cfg.name,
            "runbooks_amd_fields": dataclasses.asdict(trainer.model.cfg)}))
        # artifact-completeness marker (reference docs/design.md
        # "Buckets": reconcile logic can check completed.json in the
        # bucket after cluster re-creation)
        (Path(out_dir) / "completed.json").write_text(json.dumps(
            {"completed": True, "step": trainer.step_num}))
        print(f"trainer: done at step {trainer.step_num}; "
              f"merged servable model + artifacts in {out_dir}")
    if comm.is_dist():
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
