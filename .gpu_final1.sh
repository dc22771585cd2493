cd /root/repo
timeout 240 python -m pytest tests/test_gpu_ops.py -q 2>&1 | tail -2
python - <<'PY'
import torch, time
from runbooks_amd import ops
from runbooks_amd.ops.linear import quantize_fp8
M,N,K = 32,4096,4096
x = torch.randn(M,K,dtype=torch.bfloat16,device='cuda')
ws = [torch.randn(N,K,dtype=torch.bfloat16,device='cuda') for _ in range(8)]
q = [quantize_fp8(w) for w in ws]
for fn,name in ((lambda i: ops.ext().skinny_gemm_fp8(x,*q[i]),'fp8'),(lambda i: x@ws[i].t(),'blaslt_bf16')):
    for i in range(8): fn(i)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for r in range(6):
        for i in range(8): fn(i)
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/48
    print(f"{name}: {dt*1e6:6.1f}us")
PY
echo === train; timeout 200 python bench.py --steps 10 --warmup 3 2>gpurun_out/t9.log | tail -1 || tail -5 gpurun_out/t9.log
echo === serve; timeout 180 python bench.py --mode serve --steps 60 --warmup 5 2>gpurun_out/s9.log | tail -1 || tail -5 gpurun_out/s9.log
