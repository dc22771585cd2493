cd /root/repo
timeout 300 python -m pytest tests/test_gpu_ops.py -q -x 2>&1 | tail -3
python - <<'PY'
import torch, time
from runbooks_amd import ops
# fresh weights per iteration so W is NOT L3-resident (matches real decode)
for M,N,K in [(32,4096,4096),(32,11008,4096),(32,4096,11008),(32,32000,4096)]:
    x = torch.randn(M,K,dtype=torch.bfloat16,device='cuda')
    ws = [torch.randn(N,K,dtype=torch.bfloat16,device='cuda') for _ in range(8)]
    for fn,name in ((lambda w: ops.ext().skinny_gemm(x,w),'skinny'),(lambda w: x@w.t(),'blaslt')):
        for w in ws: fn(w)
        torch.cuda.synchronize(); t0=time.perf_counter()
        for r in range(6):
            for w in ws: fn(w)
        torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/48
        print(f"M{M} N{N} K{K} {name}: {dt*1e6:7.1f}us  {N*K*2/dt/1e12:5.2f} TB/s")
PY
echo === serve
timeout 240 python bench.py --mode serve --steps 60 --warmup 5 2>gpurun_out/s5.log | tail -1 || tail -5 gpurun_out/s5.log
echo === train
timeout 240 python bench.py --steps 10 --warmup 3 2>gpurun_out/t5.log | tail -1 || tail -5 gpurun_out/t5.log
