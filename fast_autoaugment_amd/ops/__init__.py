"""HIP extension loader.

The CDNA4 kernels live in ``csrc/`` and are built in-tree to
``fast_autoaugment_amd/ops/_C*.so`` (see setup.py at the repo root;
``python setup.py build_ext --inplace`` with PYTORCH_ROCM_ARCH=gfx950).

Policy: on a GPU machine the HIP extension is REQUIRED — any op asked to
run on a CUDA tensor without the extension raises immediately rather than
silently falling back to eager PyTorch. On CPU-only machines (CI) the pure
torch/numpy fallbacks are used, and they double as the numerics references
for the kernels.
"""
from __future__ import annotations

import importlib

import torch

_C = None
_LOAD_ERROR = None
try:
    _C = importlib.import_module("fast_autoaugment_amd.ops._C")
except Exception as e:  # pragma: no cover - exercised only when .so missing
    _LOAD_ERROR = e


def has_ext() -> bool:
    return _C is not None


def ext():
    """Return the native module, raising loudly if it should exist but doesn't."""
    if _C is None:
        raise RuntimeError(
            "fast_autoaugment_amd HIP extension (_C) is not built. "
            "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_LOAD_ERROR}"
        )
    return _C


def require_ext_for(t: torch.Tensor):
    """GPU tensors must go through the HIP kernels — no silent eager fallback."""
    if t.is_cuda:
        return ext()
    return None
