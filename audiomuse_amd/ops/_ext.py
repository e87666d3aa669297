"""Loader for the in-tree native extension (audiomuse_amd._C).

The extension is built in-tree by `setup.py build_ext --inplace` (or
`__graft_entry__.build()`), producing `audiomuse_amd/_C.*.so` which
travels with the repo snapshot to GPU boxes.

Policy (HIP_REQUIRE_NATIVE, default on): when a GPU is present the HIP
kernels MUST load — a silent fallback to eager torch on a GPU box would
invalidate benchmarks, so we fail loudly. On CPU-only hosts the torch
reference path is used and the extension is optional.
"""

from __future__ import annotations

import importlib
import threading
from typing import Any, Optional

import torch

_EXT: Optional[Any] = None
_TRIED = False
_ERR: Optional[BaseException] = None
_LOCK = threading.Lock()


def try_load() -> Optional[Any]:
    global _EXT, _TRIED, _ERR
    if not _TRIED:
        # lock + set _TRIED only after the import finishes: a sibling
        # worker thread arriving mid-import must not observe the
        # "tried, got None" state and raise (seen under --workers soak)
        with _LOCK:
            if not _TRIED:
                try:
                    _EXT = importlib.import_module("audiomuse_amd._C")
                except Exception as exc:  # noqa: BLE001
                    _ERR = exc
                    _EXT = None
                _TRIED = True
    return _EXT


def require() -> Any:
    ext = try_load()
    if ext is None:
        raise RuntimeError(
            "audiomuse_amd._C native extension is not available "
            f"(import error: {_ERR!r}). Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return ext


def native_or_none() -> Optional[Any]:
    """Return the extension, enforcing the fail-loudly policy on GPU hosts."""
    from audiomuse_amd import config

    ext = try_load()
    if ext is None and torch.cuda.is_available() and config.HIP_REQUIRE_NATIVE:
        raise RuntimeError(
            "GPU present but the audiomuse_amd._C HIP extension failed to "
            f"load ({_ERR!r}); refusing silent eager fallback. "
            "Set HIP_REQUIRE_NATIVE=0 to override (tests only)."
        )
    return ext


def native_loaded() -> bool:
    return try_load() is not None
