"""Loader for the in-tree compiled HIP extension (gfx950).

The extension is built in-tree via ``python setup.py build_ext --inplace``
(or ``__graft_entry__.build()``) so the resulting ``.so`` travels with the
repo snapshot to the GPU box.  On a CUDA/ROCm device the HIP kernels are
mandatory: if a GPU tensor reaches an op and the extension is missing we
raise instead of silently falling back to eager PyTorch.
"""

from __future__ import annotations

import importlib
import os

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("hydragnn_amd.ops._hip_ops")
    except ImportError:
        _ext = None
    return _ext


def get_extension(required: bool = False):
    ext = _load()
    if ext is None and required:
        raise RuntimeError(
            "hydragnn_amd HIP extension (_hip_ops) is not built. "
            "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH="
            "gfx950) before running on GPU. Eager fallback on GPU is "
            "disabled by design."
        )
    return ext


def has_extension() -> bool:
    return _load() is not None


def use_eager() -> bool:
    """Escape hatch for numerics parity tests only."""
    return os.environ.get("HYDRAGNN_AMD_FORCE_EAGER", "0") == "1"
