"""Loader for the gfx950 HIP extension (paddle_amd._C).

Policy (north star: kernels must actually run on GPU):
  - on a GPU box, ops REQUIRE the extension -- a missing/unbuilt _C.so
    raises instead of silently falling back to torch composites;
  - on CPU-only boxes (CI), ops use fp32 torch reference paths.
  - FLAGS_use_native_kernels=False forces the torch path (debugging).
"""
from __future__ import annotations

import torch

from . import framework

_C = None
_tried = False


def _load():
    global _C, _tried
    if _tried:
        return _C
    _tried = True
    try:
        from . import _C as mod  # in-tree _C.so
        _C = mod
    except ImportError as e:
        _C = None
        _err = e
    return _C


def get_ext(required: bool = False):
    mod = _load()
    if mod is None and required:
        raise RuntimeError(
            "paddle_amd._C (gfx950 HIP extension) is not built. Run "
            "`python csrc/build.py` (or __graft_entry__.build()). Native "
            "kernels are mandatory on GPU -- no silent eager fallback.")
    return mod


def has_ext() -> bool:
    return _load() is not None


def use_native(t: torch.Tensor) -> bool:
    """True if op dispatch should take the HIP kernel path for tensor t."""
    if not t.is_cuda:
        return False
    if not framework.get_flag("FLAGS_use_native_kernels"):
        return False
    get_ext(required=True)  # fail loudly on GPU if missing
    return True
