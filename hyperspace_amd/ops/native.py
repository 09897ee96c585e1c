"""Loader for the in-tree HIP extension.

The extension is built in-tree (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) so the ``.so`` travels with the repo snapshot.
On a GPU box the HIP path is mandatory: ops on CUDA tensors raise
``KernelUnavailableError`` if the extension failed to load — no silent
eager fallback.
"""

from __future__ import annotations

import importlib

_ext = None
_load_error = None


def _try_load():
    global _ext, _load_error
    if _ext is not None or _load_error is not None:
        return
    try:
        _ext = importlib.import_module("hyperspace_amd._hip")
    except ImportError as e:  # pragma: no cover - depends on build state
        _load_error = e


def available() -> bool:
    _try_load()
    return _ext is not None


def ext():
    _try_load()
    if _ext is None:
        from ..exceptions import KernelUnavailableError
        raise KernelUnavailableError(
            f"hyperspace_amd._hip extension not available: {_load_error}. "
            "Build it with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950).")
    return _ext
