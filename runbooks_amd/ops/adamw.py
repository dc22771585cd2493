"""Fused AdamW optimizer backed by the gfx950 HIP kernel.

One fused HBM pass per tensor per step on GPU; on CPU it runs the same
math in plain torch (used by the CPU test-suite and as the numerics
reference). Semantics match torch.optim.AdamW (decoupled weight decay,
bias correction). Moments are always fp32, also for bf16 params.
"""
from __future__ import annotations

import torch

from . import _backend


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                m, v = state["exp_avg"], state["exp_avg_sq"]
                g = p.grad
                if p.is_cuda:
                    _backend.ext().adamw_step(
                        p.data, g.contiguous(), m, v, group["lr"], beta1, beta2,
                        group["eps"], group["weight_decay"], state["step"],
                    )
                else:
                    self._ref_step(p, g, m, v, group, state["step"], beta1, beta2)
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
