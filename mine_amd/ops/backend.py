"""HIP extension loader.

The extension `_mine_hip` is built IN-TREE (mine_amd/ops/_mine_hip*.so)
by `python setup.py build_ext --inplace` / `__graft_entry__.build()` for
gfx950 only. There is no JIT fallback and no CPU shim inside the
extension: on a CUDA (ROCm) device the fused kernels are REQUIRED — if
the extension is missing, ops fail loudly rather than silently falling
back to eager PyTorch.
"""
from __future__ import annotations

import importlib
from typing import Optional

_EXT = None
_TRIED = False
_ERR: Optional[BaseException] = None


def get_extension(required: bool = False):
    global _EXT, _TRIED, _ERR
    if not _TRIED:
        _TRIED = True
        try:
            _EXT = importlib.import_module("mine_amd.ops._mine_hip")
        except Exception as e:  # noqa: BLE001
            _EXT = None
            _ERR = e
    if _EXT is None and required:
        raise RuntimeError(
            "mine_amd HIP extension (_mine_hip) is not built. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_ERR!r}")
    return _EXT


def has_extension() -> bool:
    return get_extension(required=False) is not None
