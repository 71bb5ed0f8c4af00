"""Loader for the in-tree HIP extension (draco_amd/_hip_ops*.so).

The extension is built in-tree by `setup.py build_ext --inplace` (or
`__graft_entry__.build()`), compiled by hipcc for gfx950 only.  On a GPU box the
extension is REQUIRED: ops dispatched on CUDA tensors raise if it is missing, so a
silent eager fallback can never masquerade as the native path.  Set
DRACO_ALLOW_TORCH_FALLBACK=1 to override (debug only).
"""
from __future__ import annotations

import importlib
import os

_ext = None
_tried = False
_err: Exception | None = None


def get_extension():
    global _ext, _tried, _err
    if not _tried:
        _tried = True
        try:
            _ext = importlib.import_module("draco_amd._hip_ops")
            _check_stale(_ext)
        except Exception as e:  # pragma: no cover - exercised on GPU boxes
            _ext = None
            _err = e
    return _ext


def _check_stale(ext) -> None:
    import warnings

    src = os.path.join(os.path.dirname(__file__), "csrc", "draco_kernels.hip")
    try:
        if os.path.getmtime(src) > os.path.getmtime(ext.__file__) + 1.0:
            warnings.warn(
                "draco_amd._hip_ops is OLDER than its kernel source — rebuild with "
                "`python setup.py build_ext --inplace` (stale .so gives wrong/missing ops)"
            )
    except OSError:
        pass


def require_extension():
    ext = get_extension()
    if ext is None:
        if os.environ.get("DRACO_ALLOW_TORCH_FALLBACK") == "1":
            return None
        raise RuntimeError(
            "draco_amd._hip_ops extension is not built but a CUDA tensor was passed. "
            "Build it with `python setup.py build_ext --inplace` (hipcc, gfx950). "
            f"Import error: {_err!r}"
        )
    return ext


def available() -> bool:
    return get_extension() is not None
