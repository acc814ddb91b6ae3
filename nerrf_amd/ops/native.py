"""In-tree HIP extension loader.

The extension (`nerrf_amd/_kernels*.so`) is built by `setup.py build_ext
--inplace` (driven by `__graft_entry__.build()`), targeting gfx950 only.  It
is deliberately built IN-TREE so the .so travels with repo snapshots.

Policy: on a ROCm device the native kernels are mandatory — a missing
extension raises instead of silently falling back to eager PyTorch.
"""
from __future__ import annotations

import importlib
from typing import Optional

_ext = None
_tried = False


def load_extension(required: bool = False):
    global _ext, _tried
    if _ext is not None:
        return _ext
    if not _tried or required:
        _tried = True
        try:
            _ext = importlib.import_module("nerrf_amd._kernels")
        except ImportError as e:
            _ext = None
            if required:
                raise RuntimeError(
                    "nerrf_amd._kernels HIP extension is not built but a GPU tensor "
                    "reached a fused op. Build it with `python setup.py build_ext "
                    "--inplace` (PYTORCH_ROCM_ARCH=gfx950). Refusing to fall back "
                    "to eager PyTorch on GPU."
                ) from e
    return _ext


def native_available() -> bool:
    return load_extension(required=False) is not None


def get_native(for_tensor) -> Optional[object]:
    """Return the extension when `for_tensor` lives on a GPU (required), else None."""
    if for_tensor.is_cuda:
        return load_extension(required=True)
    return None
