cd /root/repo
mkdir -p gpurun_out
python -c "import torch; print(torch.__version__, torch.cuda.get_device_name(0), torch.cuda.device_count())" 2>&1
echo "=== pytest gpu ==="
timeout 700 python -m pytest tests -m gpu -q 2>&1 | tail -25
echo "=== bench train llama2-7b ==="
timeout 420 python bench.py --gpus 1 --steps 10 --warmup 3 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
tail -1 gpurun_out/bench_train.json || tail -20 gpurun_out/bench_train.log
echo "=== bench serve llama2-7b ==="
timeout 420 python bench.py --mode serve --steps 40 --warmup 5 > gpurun_out/bench_serve.json 2> gpurun_out/bench_serve.log
tail -1 gpurun_out/bench_serve.json || tail -20 gpurun_out/bench_serve.log
echo "=== rocprof train ==="
export TMPDIR=/tmp; cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof -o train -- python /root/repo/bench.py --steps 3 --warmup 2 > $GRAFT_REPO_ROOT/gpurun_out/prof_train.log 2>&1
echo rocprof exit $?
ls $GRAFT_REPO_ROOT/gpurun_out/prof 2>&1
