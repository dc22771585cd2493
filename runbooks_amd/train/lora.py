"""LoRA adapters (Hu et al. 2021) over the transformer's linear layers.

The reference platform's fine-tune path is LoRA via the external HF
trainer image (reference examples/llama2-7b/finetuned-model.yaml,
SURVEY.md §2b "trainer image"); here it is native: frozen base weights,
trainable low-rank A/B pairs on the attention + MLP projections. The
fused fwd+bwd path (_LoRAFused) keeps every product a single hipBLASLt
call and rides both gradient merges on addmm epilogues — no separate
mul/add kernels (the eager formulation measured ~450 extra elementwise
launches per llama2-7b step; profiles/).
"""
from __future__ import annotations

import math
import os

import torch
from torch import nn

from ..parallel.tp import ColumnParallelLinear, RowParallelLinear


def _delta_add_(y, t, w, scale, w_transposed=True):
    """y += scale * t @ (w^T if w_transposed else w).

    r=16 w_transposed merges run the MFMA lora_badd_ kernel (both
    operands are natural contiguous 32x32x16 fragments; hipBLASLt runs
    these K=16 accumulates ~4x off the y-stream roofline — r33
    torch.profiler). RB_LORA_MFMA=0 reverts to in-place addmm_.
    The older scalar lora_delta_ kernel (per-lane W gather, measured
    slower) stays behind RB_LORA_KERNEL=1 for A/B work only."""
    if (os.environ.get("RB_LORA_MFMA", "1") == "1"
            and y.is_cuda and y.dtype == torch.bfloat16
            and t.shape[1] == 16 and y.shape[1] % 32 == 0
            and w.dtype == torch.bfloat16):
        from ..ops import _backend
        if _backend.use_hip(y):
            # w_transposed=False (y += t @ w, w [16, N]): transpose the
            # TINY w into the kernel's [N, 16] fragment layout — ~5 us
            # vs the 110 us hipBLASLt accumulate it replaces (r35
            # profile, [2048, 11008] dx merge).
            wk = w if w_transposed else w.t()
            _backend.ext().lora_badd_(y, t.contiguous(), wk.contiguous(),
                                      scale)
            return y
    if (os.environ.get("RB_LORA_KERNEL", "0") == "1"
            and y.is_cuda and y.dtype == torch.bfloat16
            and y.shape[1] % 8 == 0
            and t.shape[1] <= 32 and t.shape[1] % 8 == 0):
        from ..ops import _backend
        _backend.ext().lora_delta_(y, t.contiguous(), w.contiguous(),
                                   scale, w_transposed)
        return y
    return y.addmm_(t, w.t() if w_transposed else w, alpha=scale)

DEFAULT_TARGETS = ("q_proj", "k_proj", "v_proj", "o_proj",
                   "gate_proj", "up_proj", "down_proj", "fc1", "fc2")


# RB_EXPERIMENTAL_GEMM=1 routes the big frozen-weight GEMMs through the
# 8-phase 256^2 kernel (ops/csrc/gemm_train.hip) instead of hipBLASLt.
# The dgrad (dy @ W) is NN-shaped, so frozen weights get a one-time
# transposed copy (cheap next to 288 GB HBM) and run as NT(dy, W^T).
_USE_CUSTOM_GEMM = os.environ.get("RB_EXPERIMENTAL_GEMM", "0") == "1"
_WT_CACHE: dict[int, torch.Tensor] = {}


def _nt_ok(m: int, n: int, k: int) -> bool:
    return m % 256 == 0 and n % 256 == 0 and k % 128 == 0


def _nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a[M,K] @ b[N,K]^T — custom kernel when eligible, else hipBLASLt."""
    if (_USE_CUSTOM_GEMM and a.is_cuda and a.dtype == torch.bfloat16
            and _nt_ok(a.shape[0], b.shape[0], a.shape[1])):
        from ..ops import _backend
        return _backend.ext().train_gemm_nt(a.contiguous(), b.contiguous())
    return a @ b.t()


def _wt(w: torch.Tensor) -> torch.Tensor:
    wt = _WT_CACHE.get(w.data_ptr())
    if wt is None:
        wt = w.t().contiguous()
        _WT_CACHE[w.data_ptr()] = wt
    return wt


class _LoRAGroupPrep(torch.autograd.Function):
    """tcat = x @ cat(A_1..A_n)^T for sibling adapters sharing the same
    input (q/k/v; gate/up): one activation stream instead of n.

    Backward: dcat = dtcat^T x (one GEMM, x read once) sliced into the
    per-adapter dA's, and dx = dtcat @ catA (one GEMM; autograd adds it
    to the base-path dx's)."""

    @staticmethod
    def forward(ctx, x, *As):
        catA = torch.cat(As)                    # [n*r, in]
        t = x @ catA.t()                        # [T, n*r]
        ctx.save_for_backward(x, catA)
        ctx.rs = [a.shape[0] for a in As]
        return t

    @staticmethod
    def backward(ctx, dt):
        x, catA = ctx.saved_tensors
        dcat = dt.t() @ x                       # [n*r, in]
        dx = dt @ catA
        das, off = [], 0
        for r in ctx.rs:
            das.append(dcat[off:off + r])
            off += r
        return (dx, *das)


class _LoRAGroup:
    """Shared-input sibling adapters + the per-forward tcat cache (keyed
    on activation identity; dropped after every member consumed it so no
    autograd graph outlives the step)."""

    def __init__(self, members):
        self.members = members
        self._x_id = None
        self._t = None
        self._left = 0

    def t_for(self, member, x):
        if self._x_id != id(x) or self._left <= 0:
            self._t = _LoRAGroupPrep.apply(
                x, *[m.lora_a for m in self.members])
            self._x_id = id(x)
            self._left = len(self.members)
        i = self.members.index(member)
        off = sum(m.r for m in self.members[:i])
        t = self._t[:, off:off + member.r]
        self._left -= 1
        if self._left == 0:
            self._t = None
            self._x_id = None
        return t


class _LoRAFusedT(torch.autograd.Function):
    """y = x W^T + s t B^T with t precomputed by the group prep.
    dt flows back through _LoRAGroupPrep (which owns the dx merge and
    the dA GEMM for the whole group)."""

    @staticmethod
    def forward(ctx, x, w, t, b, scale):
        y = _nt(x, w)
        _delta_add_(y, t, b, scale)
        ctx.save_for_backward(w, t, b, x)
        ctx.scale = scale
        return y

    @staticmethod
    def backward(ctx, dy):
        w, t, b, x = ctx.saved_tensors
        s = ctx.scale
        if (_USE_CUSTOM_GEMM and dy.is_cuda and dy.dtype == torch.bfloat16
                and _nt_ok(dy.shape[0], w.shape[1], dy.shape[1])):
            dx = _nt(dy, _wt(w))
        else:
            dx = dy @ w
        dt = (dy @ b).mul_(s)
        db = torch.mm(dy.t(), t).mul_(s)
        return dx, None, dt, db, None


class _LoRAFused(torch.autograd.Function):
    """y = x W^T + s (x A^T) B^T with a GEMM-only backward.

    Eager autograd over the same math spends ~450 extra elementwise
    kernels per llama2-7b step on grad merges (profiles/). Here both
    merge points are addmm epilogues:
      dx = dy W  (+)= s (dy B) A     one addmm
      dA = s t2^T x,  dB = s dy^T t,  with t = x A^T, t2 = dy B
    The frozen base weight gets no wgrad at all.
    """

    @staticmethod
    def forward(ctx, x, w, a, b, scale):
        t = x @ a.t()                       # [T, r]
        # in-place addmm_: the out-of-place form first COPIES its beta
        # input (measured 448 DtoD copies x 8.2 us per train step, r11
        # torch.profiler); y is fresh from the base GEMM so accumulating
        # into it is safe
        y = _nt(x, w)
        _delta_add_(y, t, b, scale)
        ctx.save_for_backward(x, w, a, b, t)
        ctx.scale = scale
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, a, b, t = ctx.saved_tensors
        s = ctx.scale
        t2 = dy @ b                         # [T, r]
        if (_USE_CUSTOM_GEMM and dy.is_cuda and dy.dtype == torch.bfloat16
                and _nt_ok(dy.shape[0], w.shape[1], dy.shape[1])):
            dx = _nt(dy, _wt(w))
        else:
            dx = dy @ w
        _delta_add_(dx, t2, a, s, w_transposed=False)
        da = torch.mm(t2.t(), x).mul_(s)
        db = torch.mm(dy.t(), t).mul_(s)
        return dx, None, da, db, None


class LoRALinear(nn.Module):
    """Wraps a (possibly TP-sharded) linear: y = base(x) + scale * B(A(x)).

    A: [r, in_shard], B: [out_shard, r] — sharded the same way as the base
    weight so TP stays consistent; A/B are the only trainable params.
    """

    def __init__(self, base: nn.Module, r: int = 16, alpha: int = 32,
                 dropout: float = 0.0, dtype=None):
        super().__init__()
        self.base = base
        self.dropout = nn.Dropout(dropout) if dropout > 0 else None
        w = base.weight
        out_f, in_f = w.shape
        dtype = dtype or w.dtype
        self.r = r
        self.scale = alpha / r
        self.lora_a = nn.Parameter(torch.empty(r, in_f, dtype=dtype, device=w.device))
        self.lora_b = nn.Parameter(torch.zeros(out_f, r, dtype=dtype, device=w.device))
        nn.init.kaiming_uniform_(self.lora_a, a=math.sqrt(5))
        base.weight.requires_grad_(False)
        if getattr(base, "bias", None) is not None and isinstance(base.bias, nn.Parameter):
            base.bias.requires_grad_(False)

    def forward(self, x):
        base = self.base
        if self.dropout is not None and self.training:
            # dropout on the adapter input (PEFT semantics) breaks the
            # single-addmm fusion; let autograd handle the composite
            y = self.base(x)
            t = torch.nn.functional.linear(self.dropout(x), self.lora_a)
            out_f = y.shape[-1]
            return torch.addmm(y.reshape(-1, out_f),
                               t.reshape(-1, self.r), self.lora_b.t(),
                               beta=1.0, alpha=self.scale).view(y.shape)
        if (isinstance(base, nn.Linear) or
                (getattr(base, "tp", 1) == 1 and base.bias is None)) and \
                x.dim() == 2 and getattr(base, "bias", None) is None:
            group = getattr(self, "_group", None)
            if group is not None:
                # shared-input siblings (q/k/v, gate/up): the A-GEMM,
                # its dA and the t-path dx merge run ONCE per group
                t = group.t_for(self, x)
                return _LoRAFusedT.apply(x, base.weight, t, self.lora_b,
                                         self.scale)
            # fully-fused fwd+bwd path: every product is one hipBLASLt
            # call, the two gradient merges ride addmm epilogues
            return _LoRAFused.apply(x, base.weight, self.lora_a,
                                    self.lora_b, self.scale)
        y = self.base(x)
        # adapter fused into the second GEMM's epilogue:
        # y = 1*y + scale * (x A^T) B^T — addmm keeps it in hipBLASLt,
        # no materialized adapter tensor, no separate mul/add kernels.
        t = torch.nn.functional.linear(x, self.lora_a)
        out_f = y.shape[-1]
        return torch.addmm(y.reshape(-1, out_f), t.reshape(-1, self.r),
                           self.lora_b.t(), beta=1.0,
                           alpha=self.scale).view(y.shape)

    @property
    def weight(self):  # so init / inspection code keeps working
        return self.base.weight


def apply_lora(model: nn.Module, r: int = 16, alpha: int = 32,
               dropout: float = 0.0, targets=DEFAULT_TARGETS) -> list[str]:
    """Freeze the model and wrap target linears with LoRA. Returns wrapped
    module names."""
    for p in model.parameters():
        p.requires_grad_(False)
    wrapped = []
    group_on = os.environ.get("RB_LORA_GROUP", "1") != "0"
    for name, module in model.named_modules():
        for child_name, child in list(module.named_children()):
            if child_name in targets and isinstance(
                    child, (nn.Linear, ColumnParallelLinear, RowParallelLinear)):
                setattr(module, child_name,
                        LoRALinear(child, r=r, alpha=alpha, dropout=dropout))
                wrapped.append(f"{name}.{child_name}" if name else child_name)
        if group_on and dropout <= 0:
            # group shared-input siblings for the one-A-GEMM path; only
            # plain bias-free nn.Linear bases (the fused-path criteria)
            for names in (("q_proj", "k_proj", "v_proj"),
                          ("gate_proj", "up_proj")):
                ms = [getattr(module, n, None) for n in names]
                if all(isinstance(m, LoRALinear) and
                       (isinstance(m.base, nn.Linear)
                        or getattr(m.base, "tp", 1) == 1) and
                       getattr(m.base, "bias", None) is None for m in ms):
                    g = _LoRAGroup(ms)
                    for m in ms:
                        m._group = g
    if not wrapped:
        raise ValueError("apply_lora: no target modules found")
    return wrapped


def lora_state_dict(model: nn.Module) -> dict[str, torch.Tensor]:
    return {k: v for k, v in model.state_dict().items()
            if "lora_a" in k or "lora_b" in k}


def merge_lora(model: nn.Module) -> None:
    """Fold adapters into base weights (for serving a fine-tuned model)."""
    for module in model.modules():
        if isinstance(module, LoRALinear):
            with torch.no_grad():
                delta = (module.lora_b.float() @ module.lora_a.float())
                module.base.weight.add_(delta.to(module.base.weight.dtype),
                                        alpha=module.scale)
                module.lora_b.zero_()
