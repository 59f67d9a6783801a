"""Python interface to the HIP kernel extension.

On a GPU box the extension MUST be present — ops fail loudly rather than
silently falling back to torch (per-op `require()`); on CPU hosts every
caller has a torch reference path.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        import importlib

        _ext = importlib.import_module("sail_amd.ops._sail_kernels")
    except ImportError:
        try:
            import importlib

            _ext = importlib.import_module("_sail_kernels")
        except ImportError:
            _ext = None
    return _ext


def available() -> bool:
    return _load() is not None


def require():
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "sail_amd HIP kernel extension (_sail_kernels) is not built. "
            "Run `python setup.py build_ext --inplace` (gfx950).")
    return ext


def like_mask(col, pattern: str, case_insensitive=False, is_regex=False) -> Optional[torch.Tensor]:
    """Device LIKE over raw (offsets,bytes) strings. None => no kernel path
    (caller falls back on CPU; on CUDA require())."""
    if not col.is_cuda:
        return None
    ext = require()
    if is_regex or case_insensitive:
        return None  # host fallback for regex until kernel lands
    return ext.like_mask(col.offsets, col.bytes_, pattern.encode())


def string_predicate(col, kind: str, pattern: str) -> Optional[torch.Tensor]:
    if not col.is_cuda:
        return None
    ext = require()
    pat = pattern.encode()
    if kind == "contains":
        return ext.like_mask(col.offsets, col.bytes_, b"%" + pat + b"%")
    if kind == "startswith":
        return ext.like_mask(col.offsets, col.bytes_, pat + b"%")
    if kind == "endswith":
        return ext.like_mask(col.offsets, col.bytes_, b"%" + pat)
    return None
