"""HIP extension loader.

The gfx950 kernels are compiled IN-TREE (ops/_hip_ops.*.so) by
`python -m pytorch_distributed_train_amd.ops.build` (also called from
__graft_entry__.build()).

Policy: on a CUDA (ROCm) device the extension is REQUIRED — ops raise
loudly rather than falling back to eager torch, so a GPU run can never
silently measure the wrong code path. CPU tensors always use the plain
torch reference implementations in functional.py (they are the numerics
oracle for the kernels, not a production path).
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
        _ext = importlib.import_module(
            "pytorch_distributed_train_amd.ops._hip_ops")
    except ImportError as e:
        _ext = None
        _err = e
    return _ext


def has_ext() -> bool:
    return _load() is not None


def ext():
    m = _load()
    if m is None:
        raise RuntimeError(
            "pytorch_distributed_train_amd HIP extension (_hip_ops) is not "
            "built. On a GPU box this is a hard error (no silent eager "
            "fallback). Build it in-tree with: "
            "python -m pytorch_distributed_train_amd.ops.build")
    return m
