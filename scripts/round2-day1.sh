#!/usr/bin/env bash
# Round-2 day-1 GPU validation sequence (run pieces via gpurun; budget
# them — each block is one call). See NOTES-ROUND2.md for context.
set -x

# 1. Regression check of everything round 1 validated (~4 min)
python -m pytest tests -m gpu -q

# 2. Experimental 8-phase GEMM: probe -> small -> training shapes (~3 min)
RB_EXPERIMENTAL=1 python -m pytest tests/test_gpu_ops.py -q \
  -k "mfma_16x16x32 or train_gemm"

# 3. Per-kernel SoL table (~3 min); commit gpurun_out/kernels.json to profiles/
python benchmarks/kernels.py --out gpurun_out/kernels.json

# 4. TunableOp tuning run for the train GEMM shapes (~10 min); commit CSV
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop_train.csv \
  timeout 500 python bench.py --steps 5 --warmup 2
# then measure replay:
RB_TUNABLEOP_FILE=gpurun_out/tunableop_train.csv \
  timeout 300 python bench.py --steps 10 --warmup 3

# 5. Baseline benches for comparison (~4 min)
timeout 300 python bench.py --steps 10 --warmup 3
timeout 300 python bench.py --mode serve --steps 60 --warmup 5
timeout 300 python bench.py --mode serve --steps 60 --warmup 5 --serve-8bit

# 6. Late-round-1 gated features: validate then flip defaults (~2 min)
RB_EXPERIMENTAL=1 python -m pytest tests/test_gpu_ops.py -q -k 'geglu_packed or dh256'
RB_EXPERIMENTAL=1 python -m pytest tests/test_gpu_engine.py -q -k gemma -m gpu
# prefix caching on GPU (engine bookkeeping only; kernels see a block table)
RB_PREFIX_CACHE=1 python -m pytest tests/test_gpu_engine.py -q -m gpu
# then: flip RB_PREFIX_CACHE default in serve/engine.py, enable
# RB_FUSED_GEGLU, and add seq_start to attention_decode.hip for
# strict-W sliding windows (see NOTES-ROUND2.md).
