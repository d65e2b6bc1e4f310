"""Loader for the in-tree HIP extension (gfx950).

The extension is built IN-TREE (mi355ml/_hip_ext*.so) by `setup.py
build_ext --inplace` or `__graft_entry__.build()`, so the .so travels with the
repo snapshot to GPU boxes. On a CUDA(=ROCm) device the extension is
mandatory: ops raise if it is missing (no silent eager fallback) unless
config "kernels.require_hip" is False (CPU-only test runs).
"""

from __future__ import annotations

import importlib
import torch

from .. import config

_ext = None
_tried = False


def hip_ext(required: bool = True):
    """Return the compiled extension module, or raise/None."""
    global _ext, _tried
    if _ext is None and not _tried:
        _tried = True
        try:
            _ext = importlib.import_module("mi355ml._hip_ext")
        except ImportError:
            _ext = None
    if _ext is None and required:
        raise RuntimeError(
            "mi355ml HIP extension (mi355ml/_hip_ext*.so) is not built. "
            "Run `python setup.py build_ext --inplace` (or "
            "`python -c 'import __graft_entry__ as g; g.build()'`) — GPU ops "
            "never fall back to eager torch."
        )
    return _ext


def use_hip(t: torch.Tensor) -> bool:
    """True when this tensor must go through the HIP kernel path."""
    if not t.is_cuda:
        return False
    if not config.get("kernels.require_hip"):
        return hip_ext(required=False) is not None
    hip_ext(required=True)
    return True
