"""HIP kernel op surface.

``rainbowiqn_amd.ops.hip`` loads the in-tree compiled extension
(_rainbowiqn_hip.so, built by setup.py / __graft_entry__.build()).  On a GPU
box the HIP path is mandatory: if CUDA (ROCm) devices are visible and the
extension is missing, importing the accelerated path raises rather than
silently falling back to eager PyTorch (per-project rule: GPU tests must
exercise the native code).  On CPU-only machines the torch_ref implementations
are used.
"""

from __future__ import annotations

import importlib
import os

import torch

from . import torch_ref  # noqa: F401

_ext = None
_ext_err: Exception | None = None


def _load_extension():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        _ext = importlib.import_module("rainbowiqn_amd.ops._rainbowiqn_hip")
    except ImportError as e:  # pragma: no cover - depends on build state
        _ext_err = e
        _ext = None
    return _ext


def extension_available() -> bool:
    return _load_extension() is not None


def get_extension():
    ext = _load_extension()
    if ext is None:
        if torch.cuda.is_available() and os.environ.get("RIQN_ALLOW_EAGER_GPU") != "1":
            raise RuntimeError(
                "rainbowiqn_amd HIP extension (_rainbowiqn_hip) is not built but a "
                "GPU is visible -- the native path is mandatory on GPU. Build with "
                "`python setup.py build_ext --inplace` (or __graft_entry__.build()). "
                f"Import error: {_ext_err}"
            )
        raise RuntimeError(f"HIP extension unavailable (CPU-only host): {_ext_err}")
    return ext
