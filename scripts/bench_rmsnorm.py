#!/usr/bin/env python3
"""Isolated rmsnorm fwd/bwd timing on the train shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from runbooks_amd import ops

def bench(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

assert torch.cuda.is_available()
e = ops.ext()
for rows, D in [(2048, 4096), (2048, 11008)]:
    x = torch.randn(rows, D, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(D, dtype=torch.bfloat16, device="cuda")
    dy = torch.randn_like(x)
    y, ir = e.rmsnorm_fwd(x, w, 1e-5)
    t_f = bench(lambda: e.rmsnorm_fwd(x, w, 1e-5))
    t_b = bench(lambda: e.rmsnorm_bwd(x, w, dy, ir))
    gb = rows * D * 2 / 1e9
    print(f"[{rows},{D}] fwd {t_f:.1f}us ({2*gb/(t_f/1e6)/1e3:.2f} TB/s)  "
          f"bwd {t_b:.1f}us ({5*gb/(t_b/1e6)/1e3:.2f} TB/s eff)", flush=True)
