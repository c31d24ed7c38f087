"""Loader for the hand-written CDNA4 HIP kernels (gfx950).

The extension ``roko_amd/ops/_hip_ops*.so`` is built in-tree by ``setup.py
build_ext --inplace`` (or ``__graft_entry__.build()``) with
``PYTORCH_ROCM_ARCH=gfx950``. On a GPU host the model REFUSES to run without
it — a silent eager/PyTorch fallback would invalidate every benchmark number
(see the project brief), so ``require()`` raises instead.
"""

from __future__ import annotations

import importlib

# NOTE: torch is imported lazily inside the functions — feature-generation
# worker processes import roko_amd.ops._pileup through this package and must
# not pay the ~2 s torch import (measured as half the per-worker wall).

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("roko_amd.ops._hip_ops")
    except ImportError:
        _ext = None
    return _ext


def available() -> bool:
    import torch

    return torch.cuda.is_available() and _load() is not None


def require():
    import torch

    if not torch.cuda.is_available():
        raise RuntimeError("roko_amd.ops requires a ROCm GPU")
    if _load() is None:
        raise RuntimeError(
            "HIP extension roko_amd.ops._hip_ops not built — run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950); "
            "running the model on GPU without the native kernels is forbidden"
        )


def ext():
    require()
    return _ext


def pileup_ext():
    """The C++ data-path module (_pileup): BAM reader, window builder,
    alignment stats. CPU-only — no torch, no GPU required."""
    return importlib.import_module("roko_amd.ops._pileup")


def model_forward(model, x):
    """Full-model forward through the HIP kernels. Eval mode runs the fully
    fused inference path; train mode runs the differentiable path (custom
    GRU autograd + fused front)."""
    import torch

    if model.training or torch.is_grad_enabled():
        from .train import train_forward

        return train_forward(model, x)
    from .forward import roko_forward

    return roko_forward(model, x)
