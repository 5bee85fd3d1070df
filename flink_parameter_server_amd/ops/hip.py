"""Loader for the in-tree HIP/CDNA4 extension (csrc/ps_kernels.hip).

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) into ``flink_parameter_server_amd/_C*.so`` so
it travels to the GPU box with the repo snapshot.

Policy: on a machine with a visible GPU the extension is MANDATORY — any
attempt to run a device op without it raises immediately instead of
silently falling back to eager PyTorch (the fallback exists only for CPU
tests and record-mode parity).
"""
from __future__ import annotations

import importlib
from typing import Any, Optional

_ext: Optional[Any] = None
_tried = False
_err: Optional[BaseException] = None


def _load() -> Optional[Any]:
    global _ext, _tried, _err
    if not _tried:
        _tried = True
        try:
            _ext = importlib.import_module("flink_parameter_server_amd._C")
        except Exception as e:  # pragma: no cover - exercised on GPU boxes
            _err = e
            _ext = None
    return _ext


def available() -> bool:
    return _load() is not None


def ext() -> Any:
    """Return the extension module, raising loudly if it is missing."""
    m = _load()
    if m is None:
        raise RuntimeError(
            "flink_parameter_server_amd HIP extension (_C) is not built. "
            "Device ops refuse to run on a silent PyTorch fallback. "
            "Build it with `python setup.py build_ext --inplace` "
            f"(import error: {_err!r})")
    return m
