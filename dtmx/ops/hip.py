"""HIP extension loader.

The CDNA4 kernel library (dtmx/csrc/) builds in-tree to dtmx/_C*.so
(`python setup.py build_ext --inplace`, PYTORCH_ROCM_ARCH=gfx950).

Policy (fail loudly, per the MI355X-native contract): when an op runs on a
GPU tensor the extension MUST be present — there is no silent eager
fallback. CPU tensors use the torch reference path (tests, plumbing).
"""
from __future__ import annotations

import os

_ext = None
_tried = False
_err: Exception | None = None


def get_ext():
    global _ext, _tried, _err
    if not _tried:
        _tried = True
        try:
            from dtmx import _C  # built in-tree

            _ext = _C
        except ImportError as e:  # pragma: no cover
            _err = e
    return _ext


def has_ext() -> bool:
    return get_ext() is not None


def require_ext():
    ext = get_ext()
    if ext is None:
        raise RuntimeError(
            "dtmx HIP extension (dtmx/_C) is not built but a GPU tensor hit a "
            "dtmx op. Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Original import error: {_err}"
        )
    return ext
