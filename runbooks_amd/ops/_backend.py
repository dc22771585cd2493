"""Backend loader for the gfx950 HIP extension.

Policy (per the framework's no-silent-fallback rule): on a GPU box the
hand-written HIP kernels are THE compute path — if a tensor is on the GPU
and the extension is missing, ops raise instead of silently falling back
to eager PyTorch. On CPU (CI containers have no GPU) ops use plain
PyTorch fp32 reference implementations, which are also what kernel
numerics tests compare against.
"""
from __future__ import annotations

import importlib

import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        _EXT = importlib.import_module("runbooks_amd.ops._hip")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        _EXT = None
    return _EXT


def has_hip() -> bool:
    """True when the gfx950 extension is importable AND a GPU is visible."""
    return torch.cuda.is_available() and _load() is not None


def ext():
    """Return the extension module; raise loudly if we're on a GPU without it."""
    m = _load()
    if m is None:
        raise RuntimeError(
            "runbooks_amd: GPU tensor seen but the gfx950 HIP extension is not "
            "built. Run `PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext "
            f"--inplace` at the repo root. Import error: {_EXT_ERR!r}"
        )
    return m


def use_hip(*tensors: torch.Tensor) -> bool:
    """Dispatch decision for an op: any GPU tensor -> HIP path (or raise)."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if on_gpu:
        ext()  # raises if missing
        return True
    return False
