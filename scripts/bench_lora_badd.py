import sys, time, torch, json
sys.path.insert(0, ".")
from runbooks_amd import ops
for T, N in ((2048, 4096), (2048, 11008)):
    y = torch.randn(T, N, dtype=torch.bfloat16, device="cuda")
    t = torch.randn(T, 16, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, 16, dtype=torch.bfloat16, device="cuda")
    for _ in range(5): ops.ext().lora_badd_(y, t, w, 2.0)
    torch.cuda.synchronize(); e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(100): ops.ext().lora_badd_(y, t, w, 2.0)
    e1.record(); torch.cuda.synchronize()
    k_us = e0.elapsed_time(e1) * 10
    for _ in range(5): y.addmm_(t, w.t(), alpha=2.0)
    torch.cuda.synchronize(); e0.record()
    for _ in range(100): y.addmm_(t, w.t(), alpha=2.0)
    e1.record(); torch.cuda.synchronize()
    a_us = e0.elapsed_time(e1) * 10
    gb = T * N * 2 * 2 / 1e9
    print(json.dumps({"T": T, "N": N, "lora_badd_us": round(k_us, 2),
        "badd_tbs": round(gb / k_us * 1e3, 2),
        "addmm_us": round(a_us, 2), "addmm_tbs": round(gb / a_us * 1e3, 2),
        "speedup": round(a_us / k_us, 2)}))
