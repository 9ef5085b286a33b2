"""HIP extension loader.

The extension is built in-tree (maskrcnn_amd/_C*.so) by `setup.py
build_ext --inplace` or `__graft_entry__.build()` so the .so ships to the
GPU box with the repo snapshot. On a GPU box, ops MUST use the HIP path;
if a cuda tensor reaches an op and the extension is absent, we raise.
"""

from __future__ import annotations

import importlib

_ext = None
_tried = False


def load_ext():
    """Return the compiled extension module or None (CPU-only fallback)."""
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("maskrcnn_amd._C")
    except ImportError:
        _ext = None
    return _ext


def require_ext():
    """Extension required (GPU path). Raise loudly if missing."""
    ext = load_ext()
    if ext is None:
        raise RuntimeError(
            "maskrcnn_amd._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build()) "
            "before running on GPU. Refusing to fall back to eager PyTorch "
            "on a GPU tensor.")
    return ext


def has_ext() -> bool:
    return load_ext() is not None
