cd /root/repo
timeout 200 python -m pytest tests/test_gpu_ops.py -q -x -k "skinny or fast_linear" 2>&1 | tail -3
python - <<'PY'
import torch, time
from runbooks_amd import ops
torch.cuda.init()
shapes = [(32,4096,4096),(32,11008,4096),(32,4096,11008),(32,32000,4096),(1,4096,4096)]
for M,N,K in shapes:
    x = torch.randn(M,K,dtype=torch.bfloat16,device='cuda')
    w = torch.randn(N,K,dtype=torch.bfloat16,device='cuda')
    for fn, name in ((lambda: ops.ext().skinny_gemm(x,w),'skinny'), (lambda: x@w.t(),'blaslt')):
        for _ in range(10): fn()
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(50): fn()
        torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/50
        bw = N*K*2/dt/1e12
        print(f"M{M} N{N} K{K} {name}: {dt*1e6:7.1f}us  {bw:5.2f} TB/s")
PY
echo === serve
timeout 240 python bench.py --mode serve --steps 60 --warmup 5 2>gpurun_out/s4.log | tail -1 || tail -5 gpurun_out/s4.log
