cd /root/repo
timeout 200 python -m pytest tests/test_gpu_ops.py -q 2>&1 | tail -2
export TMPDIR=/tmp; cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof3 -o serve -- python /root/repo/bench.py --mode serve --steps 40 --warmup 5 > $GRAFT_REPO_ROOT/gpurun_out/pserve3.log 2>&1; echo serve-prof $?
timeout 300 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof3 -o train -- python /root/repo/bench.py --steps 4 --warmup 2 > $GRAFT_REPO_ROOT/gpurun_out/ptrain3.log 2>&1; echo train-prof $?
cd /root/repo
echo === tunableop serve
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop.csv timeout 420 python bench.py --mode serve --steps 40 --warmup 10 2>gpurun_out/s7.log | tail -1 || tail -5 gpurun_out/s7.log
tail -1 gpurun_out/pserve3.log; tail -1 gpurun_out/ptrain3.log
