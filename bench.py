#!/usr/bin/env python3
"""Flagship benchmark for the driver contract.

Default (--mode both) emits BOTH BASELINE.json headline metrics as
sequential JSON lines: the llama2-7b LoRA fine-tune step (DP over
RCCL/xGMI, "finetune samples/sec") first, then "Server tokens/sec"
(paged-KV batch-32 decode on the same model), bf16, synthetic token
data, random-init weights. The reference publishes no numbers; this
bench ESTABLISHES the baseline. --mode train / serve
select a single metric.

Launch (driver):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

Scaling: train is weak (per-GPU micro-batch fixed, DP over RCCL);
serve under torchrun is strong (TP: rank 0 drives the engine, worker
ranks follow the broadcast protocol). Rank 0 prints one JSON line per
metric.
"""
from __future__ import annotations

import argparse
import json
import os
import time

# hipBLASLt TunableOp replay: profiles/tunableop_train.csv holds the
# offline-tuned GEMM selections for the train step shapes (measured
# +7%: 105 -> 98 ms/step, gpurun r8). Torch resolves the filename by
# inserting the DEVICE ordinal before the extension, and under torchrun
# rank r runs on device r — so stage a per-ordinal copy in /tmp.
# RB_TUNABLEOP_FILE overrides; RB_TUNABLEOP=0 disables.
def _setup_tunableop():
    if os.environ.get("RB_TUNABLEOP", "1") != "1":
        return
    src = os.environ.get("RB_TUNABLEOP_FILE") or os.path.join(
        os.path.dirname(os.path.abspath(__file__)),
        "profiles", "tunableop_train.csv")
    if not os.path.exists(src):
        return
    import shutil
    base = f"/tmp/rb_tunableop_{os.getpid()}.csv"
    stem, ext = os.path.splitext(base)
    for dev in range(8):
        shutil.copy(src, f"{stem}{dev}{ext}")
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", base)


_setup_tunableop()

import torch

from runbooks_amd.parallel import comm


def _sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def _max_over_ranks(x: float) -> float:
    if comm.is_dist():
        t = torch.tensor([x], dtype=torch.float64)
        if torch.cuda.is_available():
            t = t.cuda()
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        return float(t.item())
    return x


def bench_train(args) -> dict:
    from runbooks_amd.models import get_config
    from runbooks_amd.train import TrainConfig, Trainer
    from runbooks_amd.train.data import SyntheticTokens

    args.seq_len = min(args.seq_len, get_config(args.model).max_seq_len - 1)
    cfg = TrainConfig(model=args.model, seq_len=args.seq_len,
                      micro_batch=args.micro_batch,
                      num_train_steps=args.warmup + args.steps,
                      dtype="bfloat16" if torch.cuda.is_available() else "float32",
                      save_steps=0, seed=17)
    trainer = Trainer(cfg)
    vocab = trainer.model.cfg.vocab_size
    ds = SyntheticTokens(vocab, args.seq_len + 1, n=4096, seed=comm.rank())
    batches = [torch.stack([ds[i * args.micro_batch + j]
                            for j in range(args.micro_batch)])
               for i in range(min(8, 4096 // args.micro_batch))]

    for i in range(args.warmup):
        trainer.train_step(batches[i % len(batches)])
    comm.barrier()
    _sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        trainer.train_step(batches[i % len(batches)])
    comm.barrier()
    _sync()
    dt = _max_over_ranks(time.perf_counter() - t0)

    n = comm.world_size()
    samples = args.steps * args.micro_batch * n
    return {
        "metric": "finetune_samples_per_sec",
        "value": samples / dt,
        "unit": "samples/sec",
        "n_gpus": n,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if torch.cuda.is_available() else "float32",
        "data": "synthetic",
        "config": {"model": args.model, "global_batch": args.micro_batch * n,
                   "seq_len": args.seq_len,
                   "parallelism": f"dp{n}", "lora_r": cfg.lora_r},
    }


def bench_serve(args) -> dict:
    from runbooks_amd.serve import Engine

    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    eng = Engine(args.model, dtype=dtype,
                 kv_blocks=None if torch.cuda.is_available() else 4096,
                 max_batch=args.serve_batch, seed=17,
                 load_in_8bit=args.serve_8bit,
                 kv_fp8=args.serve_fp8kv or None)
    if comm.is_dist() and comm.world_size() > 1:
        # TP serving: rank 0 drives the engine + measures; workers follow
        # the broadcast protocol until shutdown (serve/tp_worker.py).
        if comm.rank() != 0:
            from runbooks_amd.serve.tp_worker import worker_loop
            worker_loop(eng)
            return {}
    vocab = eng.cfg.vocab_size
    g = torch.Generator().manual_seed(17)
    # leave decode headroom within the model's positional range (matters
    # for the tiny CPU default, max_seq_len=128)
    prompt_len = min(args.prompt_len, eng.cfg.max_seq_len // 2)

    def new_req():
        prompt = torch.randint(0, vocab, (prompt_len,), generator=g).tolist()
        eng.submit(prompt, max_new_tokens=1 << 30)  # run until bench ends

    for _ in range(args.serve_batch):
        new_req()
    # warmup: prefill everyone + W decode steps
    for _ in range(args.serve_batch + args.warmup):
        eng.step()
    assert len(eng.running) == args.serve_batch
    _sync()
    t0 = time.perf_counter()
    tokens = 0
    for _ in range(args.steps):
        before = sum(len(r.output_ids) for r in eng.running)
        eng.step()
        after = sum(len(r.output_ids) for r in eng.running)
        tokens += after - before
    _sync()
    dt = time.perf_counter() - t0
    if comm.is_dist() and comm.world_size() > 1:
        from runbooks_amd.serve.tp_worker import broadcast_shutdown
        broadcast_shutdown(torch.device(eng.device))
    return {
        "metric": "serve_tokens_per_sec",
        "value": tokens / dt,
        "unit": "tokens/sec",
        "n_gpus": comm.world_size(),
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1e3,
        "higher_is_better": True,
        # TP serving keeps total work fixed as ranks grow
        "scaling": "strong" if comm.world_size() > 1 else "weak",
        "vs_baseline": None,
        "dtype": (("bf16-act+fp8-w" if args.serve_8bit else
                   "bf16+fp8-kv" if args.serve_fp8kv else "bf16")
                  if torch.cuda.is_available() else "float32"),
        "data": "synthetic",
        "config": {"model": args.model, "global_batch": args.serve_batch,
                   "seq_len": prompt_len,
                   "parallelism": f"tp{comm.world_size()}"},
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--mode", choices=["both", "train", "serve"],
                   default="both")
    p.add_argument("--model", default=None)
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--micro-batch", type=int, default=4)
    p.add_argument("--serve-batch", type=int, default=32)
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--serve-8bit", action="store_true",
                   help="fp8-e4m3 weight-only decode (NOT the headline "
                        "config; reported with dtype=bf16-act+fp8-w)")
    p.add_argument("--serve-fp8kv", action="store_true",
                   help="fp8-e4m3 KV cache (2x KV capacity; NOT the "
                        "headline config — reported with "
                        "dtype=bf16+fp8-kv)")
    args = p.parse_args()
    if args.model is None:
        if torch.cuda.is_available():
            args.model = "llama2-7b"
        else:
            # CPU rehearsal: pick a tiny model whose heads split across
            # the requested TP degree (driver scale run goes to ws=8)
            ws = int(os.environ.get("WORLD_SIZE", "1"))
            args.model = "tiny-llama" if ws <= 2 else "tiny-llama-8h"

    comm.init_from_env()
    torch.manual_seed(17)

    # Default emits BOTH headline metrics sequentially (BASELINE.json:
    # "Server tokens/sec + finetune samples/sec") — one JSON line each,
    # finetune first, serve last.
    results = []
    if args.mode in ("both", "train"):
        results.append(bench_train(args))
    if args.mode in ("both", "serve"):
        if args.mode == "both":
            # release train model/optimizer before the engine allocates
            # weights + KV cache
            import gc
            gc.collect()
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
        results.append(bench_serve(args))
    if comm.rank() == 0:
        for r in results:
            if r:
                print(json.dumps(r), flush=True)
    if comm.is_dist():
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
