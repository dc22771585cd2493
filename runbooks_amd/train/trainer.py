"""The fine-tune loop: LoRA (or full) training with DP over RCCL.

Native replacement for the reference's external HF trainer image
(SURVEY.md §2b "trainer image"): params arrive via PARAM_* env /
params.json (utils.params), data from /content/data, checkpoints to
/content/artifacts.
"""
from __future__ import annotations

import time
from dataclasses import dataclass

import torch

from ..models import build_model
from .. import ops
from ..ops import FusedAdamW
from ..parallel import DataParallel, comm
from . import checkpoint as ckpt_mod
from .data import SyntheticTokens, data_loader
from .lora import apply_lora, lora_state_dict


@dataclass
class TrainConfig:
    model: str = "llama2-7b"
    seq_len: int = 512
    micro_batch: int = 4
    lr: float = 2e-4
    lr_scheduler: str = "constant"      # constant | cosine | linear
    warmup_steps: int = 0
    weight_decay: float = 0.0
    num_train_steps: int = 100
    save_steps: int = 0                 # 0 = no checkpoints
    grad_accum_steps: int = 1
    lora_r: int = 16
    lora_alpha: int = 32
    lora_dropout: float = 0.0
    full_finetune: bool = False
    grad_checkpointing: bool = False    # recompute block activations
    eval_steps: int = 0                 # 0 = no evaluation
    grad_clip: float = 1.0
    dtype: str = "bfloat16"
    seed: int = 0
    output_dir: str = "/content/artifacts"


class Trainer:
    def __init__(self, cfg: TrainConfig, device=None):
        self.cfg = cfg
        self.device = device if device is not None else (
            f"cuda:{comm.local_rank()}" if torch.cuda.is_available() else "cpu")
        dtype = getattr(torch, cfg.dtype)
        model = build_model(cfg.model, dtype=dtype, tp=1, seed=cfg.seed,
                            device=self.device)
        if not cfg.full_finetune:
            apply_lora(model, r=cfg.lora_r, alpha=cfg.lora_alpha,
                       dropout=cfg.lora_dropout)
        if cfg.grad_checkpointing:
            model.enable_grad_checkpointing()
        self.ddp = DataParallel(model)
        self.model = model
        trainable = [p for p in model.parameters() if p.requires_grad]
        self.optimizer = FusedAdamW(trainable, lr=cfg.lr,
                                    weight_decay=cfg.weight_decay)
        self.step_num = 0

    # -- core step -------------------------------------------------------------
    def train_step(self, tokens: torch.Tensor, sync: bool = True) -> float:
        """tokens [B, S+1] (inputs + shifted labels). Returns loss.
        sync=False runs a gradient-accumulation micro-batch: grads pile
        into the flat buckets, no all-reduce, no optimizer step."""
        inputs = tokens[:, :-1].to(self.device)
        labels = tokens[:, 1:].to(self.device)
        self.ddp.start_microbatch(sync=sync)
        logits = self.ddp(inputs)
        scale = 1.0 / max(1, self.cfg.grad_accum_steps)
        loss = ops.cross_entropy(logits, labels)
        (loss * scale if scale != 1.0 else loss).backward()
        if not sync:
            return float(loss.detach())
        self.ddp.finish_backward()
        if self.cfg.grad_clip > 0:
            # grads are views into the DDP flat buckets — clip the few flat
            # buffers instead of hundreds of per-tensor norms.
            bufs = self.ddp.grad_buffers()
            norms = torch._foreach_norm(bufs)
            total = torch.linalg.vector_norm(torch.stack(norms))
            scale = self.cfg.grad_clip / (total + 1e-6)
            if float(scale) < 1.0:
                torch._foreach_mul_(bufs, scale)
        self._apply_lr_schedule()
        self.optimizer.step()
        self.ddp.zero_grad()
        self.step_num += 1
        return float(loss.detach())

    def _apply_lr_schedule(self):
        """HF-TrainingArguments-style warmup + decay on the base lr."""
        import math
        cfg = self.cfg
        step = self.step_num + 1
        scale = 1.0
        if cfg.warmup_steps > 0 and step <= cfg.warmup_steps:
            scale = step / cfg.warmup_steps
        elif cfg.lr_scheduler in ("cosine", "linear"):
            total = max(1, cfg.num_train_steps - cfg.warmup_steps)
            done = min(total, step - cfg.warmup_steps)
            frac = done / total
            scale = (0.5 * (1.0 + math.cos(math.pi * frac))
                     if cfg.lr_scheduler == "cosine" else 1.0 - frac)
        for g in self.optimizer.param_groups:
            g["lr"] = cfg.lr * scale

    # -- full loop -------------------------------------------------------------
    @torch.no_grad()
    def evaluate(self, dataset, max_batches: int = 16) -> float:
        """Mean cross-entropy over (up to) max_batches of an eval split —
        the eval_loss the reference's HF trainer images log."""
        self.model.eval()
        loader = data_loader(dataset, self.cfg.micro_batch,
                             rank=comm.rank(), world=comm.world_size(),
                             seed=self.cfg.seed + 1)
        total, n = 0.0, 0
        for batch in loader:
            inputs = batch[:, :-1].to(self.device)
            labels = batch[:, 1:].to(self.device)
            total += float(ops.cross_entropy(self.model(inputs), labels))
            n += 1
            if n >= max_batches:
                break
        self.model.train()
        if comm.world_size() > 1:
            # must live on self.device: the RCCL backend only reduces
            # CUDA tensors (a CPU tensor here crashes multi-GPU eval)
            t = torch.tensor([total, float(n)], device=self.device)
            torch.distributed.all_reduce(t)
            total, n = float(t[0]), int(t[1])
        return total / max(1, n)

    def fit(self, dataset=None, eval_dataset=None, log_every: int = 10):
        cfg = self.cfg
        if dataset is None:
            dataset = SyntheticTokens(self.model.cfg.vocab_size, cfg.seq_len + 1)
        loader = data_loader(dataset, cfg.micro_batch, rank=comm.rank(),
                             world=comm.world_size(), seed=cfg.seed)
        it = iter(loader)
        t0 = time.time()
        while self.step_num < cfg.num_train_steps:
            try:
                batch = next(it)
            except StopIteration:
                it = iter(loader)
                batch = next(it)
            for _ in range(self.cfg.grad_accum_steps - 1):
                self.train_step(batch, sync=False)
                try:
                    batch = next(it)
                except StopIteration:
                    it = iter(loader)
                    batch = next(it)
            from ..utils.trace import get_tracer
            tr = get_tracer()
            if tr:
                with tr.span("train_step", step=self.step_num):
                    loss = self.train_step(batch)
            else:
                loss = self.train_step(batch)
            if cfg.eval_steps and eval_dataset is not None and \
                    self.step_num % cfg.eval_steps == 0:
                ev = self.evaluate(eval_dataset)
                if comm.rank() == 0:
                    print(f"step {self.step_num} eval_loss {ev:.4f}",
                          flush=True)
            if cfg.save_steps and self.step_num % cfg.save_steps == 0:
                self.save()
            if comm.rank() == 0 and self.step_num % log_every == 0:
                dt = time.time() - t0
                sps = self.step_num * cfg.micro_batch * comm.world_size() / max(dt, 1e-9)
                print(f"step {self.step_num} loss {loss:.4f} "
                      f"samples/sec {sps:.2f}", flush=True)
        if cfg.save_steps:
            self.save()
        from ..utils.trace import dump_global
        dump_global()

    def save(self):
        if comm.rank() != 0:
            return
        state = (self.model.state_dict() if self.cfg.full_finetune
                 else lora_state_dict(self.model))
        ckpt_mod.save_checkpoint(self.cfg.output_dir, self.step_num, state,
                                 model_cfg=self.model.cfg,
                                 optim_state=self.optimizer.state_dict())

    def resume(self) -> bool:
        """Resume model weights, optimizer moments, and the step counter
        from the latest checkpoint in output_dir (the bucket-mounted
        /content/artifacts in-cluster — artifacts survive cluster
        re-creation via the deterministic bucket path)."""
        latest = ckpt_mod.latest_checkpoint(self.cfg.output_dir)
        if latest is None:
            return False
        state, optim, step = ckpt_mod.load_checkpoint(latest)
        self.model.load_state_dict(state, strict=False)
        if optim is not None:
            try:
                self.optimizer.load_state_dict(optim)
            except (ValueError, KeyError):
                pass  # param set changed; train on with fresh moments
        self.step_num = step
        if comm.is_dist():
            # ranks > 0 load the same weights via rank-0 broadcast
            self.ddp._broadcast_params()
        return True
