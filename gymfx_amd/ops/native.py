"""Loader for the in-tree HIP extension (_gymfx_hip).

The extension is built IN-TREE (setup.py build_ext --inplace /
__graft_entry__.build()) so the .so travels with the repo snapshot to GPU
boxes.  On a machine with a GPU the native path is REQUIRED: ops fail
loudly rather than silently falling back to eager torch (the torch path in
envs/reference_step.py is the CPU oracle, not a production fallback).
"""
from __future__ import annotations

import importlib
from typing import Any, Optional

import torch

_ext: Optional[Any] = None
_tried = False


def load() -> Optional[Any]:
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("gymfx_amd.ops._gymfx_hip")
    except ImportError:
        _ext = None
    return _ext


def available() -> bool:
    return load() is not None


def require():
    """Return the native extension module; raise loudly when missing."""
    ext = load()
    if ext is None:
        raise RuntimeError(
            "gymfx_amd HIP extension (_gymfx_hip) is not built. Build it "
            "in-tree with: python setup.py build_ext --inplace "
            "(PYTORCH_ROCM_ARCH=gfx950). The eager torch path is only a CPU "
            "oracle and is not used on GPU."
        )
    return ext


__all__ = ["load", "available", "require"]
