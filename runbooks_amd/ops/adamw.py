"""Fused AdamW optimizer backed by the gfx950 HIP kernels.

GPU path: ONE multi-tensor launch per (dtype-pair) group per step
(adamw_step_multi, csrc/adamw.hip) over a cached chunk table — grads are
DDP bucket views and moments are allocated once, so the device pointers
are stable across steps and the table is built once. Falls back to the
per-tensor fused kernel when a pointer changes, and to plain torch math
on CPU (the numerics reference). Semantics match torch.optim.AdamW
(decoupled weight decay, bias correction); moments are always fp32.
"""
from __future__ import annotations

import torch

from . import _backend


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._mt_cache = None  # (ptr_signature, [(table, nchunks, p32, g32)])

    def _init_state(self, p):
        state = self.state[p]
        state["step"] = 0
        state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
        state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
        return state

    def _build_mt_tables(self, group):
        """Chunk tables per (p32, g32) combo for one param group."""
        chunk = int(_backend.ext().adamw_mt_chunk_elems())
        combos: dict[tuple, list] = {}
        sig = []
        for p in group["params"]:
            if p.grad is None:
                continue
            g = p.grad
            if not g.is_contiguous():
                return None, None  # fall back to per-tensor path
            state = self.state[p]
            key = (p.dtype == torch.float32, g.dtype == torch.float32)
            rows = combos.setdefault(key, [])
            n = p.numel()
            pp, gp = p.data_ptr(), g.data_ptr()
            mp = state["exp_avg"].data_ptr()
            vp = state["exp_avg_sq"].data_ptr()
            sig.append((pp, gp))
            off = 0
            while off < n:
                cn = min(chunk, n - off)
                e = p.element_size()
                ge = g.element_size()
                rows.append((pp + off * e, gp + off * ge, mp + off * 4,
                             vp + off * 4, cn))
                off += cn
        dev = group["params"][0].device
        tables = []
        for (p32, g32), rows in combos.items():
            t = torch.tensor(rows, dtype=torch.int64).to(dev,
                                                         non_blocking=True)
            tables.append((t, len(rows), p32, g32))
        return sig, tables

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            # lazy state init + shared step counter
            step_num = None
            uniform_step = True
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    self._init_state(p)
                state["step"] += 1
                if step_num is None:
                    step_num = state["step"]
                elif state["step"] != step_num:
                    uniform_step = False

            if step_num is None:
                continue

            if uniform_step and group["params"][0].is_cuda:
                sig = [(p.data_ptr(), p.grad.data_ptr())
                       for p in group["params"] if p.grad is not None]
                if self._mt_cache is None or self._mt_cache[0] != sig:
                    built_sig, tables = self._build_mt_tables(group)
                    self._mt_cache = (built_sig, tables) if tables else None
                if self._mt_cache is not None:
                    for table, nchunks, p32, g32 in self._mt_cache[1]:
                        _backend.ext().adamw_step_multi(
                            table, nchunks, p32, g32, group["lr"], beta1,
                            beta2, group["eps"], group["weight_decay"],
                            step_num)
                    continue

            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                m, v = state["exp_avg"], state["exp_avg_sq"]
                g = p.grad
                if p.is_cuda:
                    _backend.ext().adamw_step(
                        p.data, g.contiguous(), m, v, group["lr"], beta1,
                        beta2, group["eps"], group["weight_decay"],
                        state["step"])
                else:
                    self._ref_step(p, g, m, v, group, state["step"],
                                   beta1, beta2)
        return loss

    @staticmethod
    def _ref_step(p, g, m, v, group, step, beta1, beta2):
        gf = g.float()
        pf = p.data.float()
        m.mul_(beta1).add_(gf, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
        bc1 = 1 - beta1 ** step
        bc2 = 1 - beta2 ** step
        denom = (v / bc2).sqrt_().add_(group["eps"])
        pf -= group["lr"] * ((m / bc1) / denom + group["weight_decay"] * pf)
        p.data.copy_(pf.to(p.dtype))
