#!/usr/bin/env python3
"""Per-kernel microbenchmarks for the gfx950 HIP library.

Run on an MI355X box (one gpurun call):

    python benchmarks/kernels.py [--out gpurun_out/kernels.json]

Each row: kernel, shape, wall us (median of 30 after 10 warmup via CUDA
events), achieved GB/s against its minimal-traffic model, and the
speed-of-light fraction at 6.3 TB/s (the measured HBM rate,
MI355X_MICROARCH.md). Weight-streaming benches use FRESH tensors per
iteration so the 256 MB Infinity Cache cannot fake the rate.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from runbooks_amd import ops

HBM = 6.3e12  # achievable B/s


def timeit(fn, iters=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    times = []
    start, end = torch.cuda.Event(True), torch.cuda.Event(True)
    for _ in range(iters):
        start.record()
        fn()
        end.record()
        torch.cuda.synchronize()
        times.append(start.elapsed_time(end) * 1e3)  # us
    times.sort()
    return times[len(times) // 2]


def row(name, shape, us, bytes_moved, flops=0):
    gbs = bytes_moved / (us * 1e-6) / 1e9
    return {"kernel": name, "shape": shape, "us": round(us, 2),
            "GB/s": round(gbs, 1), "sol": round(gbs * 1e9 / HBM, 3),
            "TF/s": round(flops / (us * 1e-6) / 1e12, 1) if flops else None}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default=None)
    args = p.parse_args()
    assert torch.cuda.is_available() and ops.has_hip()
    dev = "cuda"
    bf16 = torch.bfloat16
    out = []

    # rmsnorm fwd/bwd [2048, 4096]
    x = torch.randn(2048, 4096, dtype=bf16, device=dev)
    w = torch.randn(4096, dtype=bf16, device=dev)
    us = timeit(lambda: ops.rmsnorm(x, w, 1e-5))
    out.append(row("rmsnorm_fwd", "2048x4096", us, 2 * x.numel() * 2))
    y, inv = ops.ext().rmsnorm_fwd(x, w, 1e-5)
    dy = torch.randn_like(x)
    us = timeit(lambda: ops.ext().rmsnorm_bwd(x, w, dy, inv))
    out.append(row("rmsnorm_bwd", "2048x4096", us, 5 * x.numel() * 2))

    # rope [2048, 32, 128]
    q = torch.randn(2048, 32, 128, dtype=bf16, device=dev)
    cos, sin = ops.rope_tables(128, 4096, device=dev)
    pos = torch.arange(2048, dtype=torch.int32, device=dev) % 512
    us = timeit(lambda: ops.rope(q, cos, sin, pos))
    out.append(row("rope_fwd", "2048x32x128", us, 2 * q.numel() * 2))

    # swiglu fwd [2048, 11008]
    g = torch.randn(2048, 11008, dtype=bf16, device=dev)
    u = torch.randn_like(g)
    us = timeit(lambda: ops.swiglu(g, u))
    out.append(row("swiglu_fwd", "2048x11008", us, 3 * g.numel() * 2))

    # packed GLU epilogues [2048, 2*11008] (swiglu validated; geglu gated)
    yp = torch.randn(2048, 2 * 11008, dtype=bf16, device=dev)
    us = timeit(lambda: ops.ext().swiglu_packed(yp))
    out.append(row("swiglu_packed", "2048x2x11008", us,
                   1.5 * yp.numel() * 2))
    us = timeit(lambda: ops.ext().geglu_packed(yp))
    out.append(row("geglu_packed", "2048x2x11008", us,
                   1.5 * yp.numel() * 2))

    # cross-entropy fwd+bwd [2048, 32000]
    logits = torch.randn(4, 512, 32000, dtype=bf16, device=dev,
                         requires_grad=True)
    tgt = torch.randint(0, 32000, (4, 512), device=dev)

    def ce():
        loss = ops.cross_entropy(logits, tgt)
        loss.backward()
        logits.grad = None
    us = timeit(ce)
    out.append(row("cross_entropy_fwd+bwd", "2048x32000", us,
                   3 * logits.numel() * 2))

    # flash prefill / training fwd [4, 512, 32, 128]
    qq = torch.randn(4, 512, 32, 128, dtype=bf16, device=dev)
    kk, vv = torch.randn_like(qq), torch.randn_like(qq)
    fl = 4 * 4 * 32 * 512 * 512 * 128 / 2  # causal half
    us = timeit(lambda: ops.flash_prefill(qq, kk, vv))
    out.append(row("flash_prefill", "b4 s512 h32 d128", us,
                   4 * qq.numel() * 2, flops=fl))
    qg = qq.clone().requires_grad_(True)
    kg = kk.clone().requires_grad_(True)
    vg = vv.clone().requires_grad_(True)
    dyy = torch.randn_like(qq)

    def fabwd():
        o = ops.causal_attention(qg, kg, vg)
        o.backward(dyy)
        qg.grad = kg.grad = vg.grad = None
    us = timeit(fabwd, iters=15)
    out.append(row("flash_fwd+bwd", "b4 s512 h32 d128", us,
                   10 * qq.numel() * 2, flops=3.5 * fl))

    # paged decode b32 len512
    BS, B, L = 16, 32, 512
    nb = B * (L // BS) + 8
    kc = torch.randn(nb, 32, BS, 128, dtype=bf16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.arange(B * (L // BS), dtype=torch.int32,
                      device=dev).reshape(B, -1)
    sl = torch.full((B,), L, dtype=torch.int32, device=dev)
    qd = torch.randn(B, 32, 128, dtype=bf16, device=dev)
    us = timeit(lambda: ops.paged_decode(qd, kc, vc, bt, sl))
    out.append(row("paged_decode", f"b{B} len{L} h32 d128", us,
                   2 * B * 32 * L * 128 * 2))

    # gemma-width decode (Dh=256 instantiation; first validated round 2)
    kc2 = torch.randn(nb // 2, 16, BS, 256, dtype=bf16, device=dev)
    vc2 = torch.randn_like(kc2)
    bt2 = torch.arange(B // 2 * (L // BS), dtype=torch.int32,
                       device=dev).reshape(B // 2, -1)
    sl2 = torch.full((B // 2,), L, dtype=torch.int32, device=dev)
    qd2 = torch.randn(B // 2, 16, 256, dtype=bf16, device=dev)
    us = timeit(lambda: ops.paged_decode(qd2, kc2, vc2, bt2, sl2))
    out.append(row("paged_decode_dh256", f"b{B//2} len{L} h16 d256", us,
                   2 * (B // 2) * 16 * L * 256 * 2))

    # MFMA GQA decode (llama2-70b shape) over the transposed-V layout
    kc3 = torch.randn(nb, 8, BS, 128, dtype=bf16, device=dev)
    vc3 = torch.randn_like(kc3).permute(0, 1, 3, 2).contiguous()
    qd3 = torch.randn(B, 64, 128, dtype=bf16, device=dev)
    us = timeit(lambda: ops.paged_decode(qd3, kc3, vc3, bt, sl))
    out.append(row("paged_decode_mfma_gqa", f"b{B} len{L} 64/8 d128", us,
                   2 * B * 8 * L * 128 * 2))

    # MFMA LoRA B-merge vs the hipBLASLt K=16 accumulate
    yb = torch.randn(2048, 11008, dtype=bf16, device=dev)
    tb = torch.randn(2048, 16, dtype=bf16, device=dev)
    wb = torch.randn(11008, 16, dtype=bf16, device=dev)
    us = timeit(lambda: ops.ext().lora_badd_(yb, tb, wb, 2.0))
    out.append(row("lora_badd", "2048x11008 r16", us, 2 * yb.numel() * 2))
    us = timeit(lambda: yb.addmm_(tb, wb.t(), alpha=2.0))
    out.append(row("blaslt_lora_addmm", "2048x11008 r16", us,
                   2 * yb.numel() * 2))

    # decode GEMMs: fresh weights per call (no L3 reuse)
    for N, K in [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008),
                 (32000, 4096)]:
        xs = torch.randn(32, K, dtype=bf16, device=dev)
        ws = [torch.randn(N, K, dtype=bf16, device=dev) for _ in range(10)]
        i = [0]

        def blas():
            i[0] = (i[0] + 1) % 10
            return xs @ ws[i[0]].t()
        us = timeit(blas)
        out.append(row(f"blaslt_decode", f"32x{N}x{K}", us, N * K * 2,
                       flops=2 * 32 * N * K))
        def skinny():
            i[0] = (i[0] + 1) % 10
            return ops.ext().skinny_gemm(xs, ws[i[0]])
        us = timeit(skinny)
        out.append(row(f"skinny_decode", f"32x{N}x{K}", us, N * K * 2,
                       flops=2 * 32 * N * K))
        # v2 weight-stream kernel over pre-swizzled operands (the
        # shipped decode path)
        sws = [ops.ext().decode_swizzle_w(w) for w in ws]
        sx = ops.ext().decode_swizzle_x(xs)

        def dgv2():
            i[0] = (i[0] + 1) % 10
            return ops.ext().decode_gemm(sx, sws[i[0]], 32, N, K)
        us = timeit(dgv2)
        out.append(row(f"decode_gemm_v2", f"32x{N}x{K}", us, N * K * 2,
                       flops=2 * 32 * N * K))
        del sws

    # fp8 decode GEMM
    from runbooks_amd.ops.linear import quantize_fp8
    xs = torch.randn(32, 4096, dtype=bf16, device=dev)
    qs = [quantize_fp8(torch.randn(12288, 4096, dtype=bf16, device=dev))
          for _ in range(10)]
    i = [0]

    def fp8():
        i[0] = (i[0] + 1) % 10
        return ops.ext().skinny_gemm_fp8(xs, *qs[i[0]])
    us = timeit(fp8)
    out.append(row("skinny_fp8_decode", "32x12288x4096", us, 12288 * 4096,
                   flops=2 * 32 * 12288 * 4096))

    # train GEMM shapes (hipBLASLt, fresh weights)
    for M, N, K in [(2048, 4096, 4096), (2048, 11008, 4096),
                    (2048, 4096, 11008)]:
        xs = torch.randn(M, K, dtype=bf16, device=dev)
        ws = [torch.randn(N, K, dtype=bf16, device=dev) for _ in range(6)]
        i = [0]

        def tg():
            i[0] = (i[0] + 1) % 6
            return xs @ ws[i[0]].t()
        us = timeit(tg)
        out.append(row("blaslt_train", f"{M}x{N}x{K}", us,
                       (M * K + N * K + M * N) * 2, flops=2 * M * N * K))

    # fused AdamW multi-tensor: 224 LoRA-sized tensors
    ps = [torch.nn.Parameter(torch.randn(16 * 4096, dtype=bf16, device=dev))
          for _ in range(224)]
    opt = ops.FusedAdamW(ps, lr=1e-3)
    for pp in ps:
        pp.grad = torch.randn_like(pp)
    us = timeit(opt.step)
    total = sum(pp.numel() for pp in ps)
    out.append(row("adamw_multi_tensor", "224x64K", us, total * (2 + 2 + 16)))

    for r in out:
        print(f"{r['kernel']:>22} {r['shape']:>16} {r['us']:>9.1f}us "
              f"{r['GB/s']:>7.1f} GB/s  sol={r['sol']:<5}"
              f" {('%.0f TF/s' % r['TF/s']) if r['TF/s'] else ''}")
    if args.out:
        with open(args.out, "w") as f:
            json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
