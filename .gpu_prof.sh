cd /root/repo
mkdir -p gpurun_out
timeout 200 python -m pytest tests/test_gpu_ops.py -q 2>&1 | tail -2
export TMPDIR=/tmp; cd /tmp
timeout 360 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof2 -o train -- python /root/repo/bench.py --steps 4 --warmup 2 > $GRAFT_REPO_ROOT/gpurun_out/prof_train2.log 2>&1
echo train rocprof $?
timeout 360 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof2 -o serve -- python /root/repo/bench.py --mode serve --steps 60 --warmup 5 > $GRAFT_REPO_ROOT/gpurun_out/prof_serve2.log 2>&1
echo serve rocprof $?
tail -1 $GRAFT_REPO_ROOT/gpurun_out/prof_train2.log
tail -1 $GRAFT_REPO_ROOT/gpurun_out/prof_serve2.log
ls $GRAFT_REPO_ROOT/gpurun_out/prof2
