"""Loader for the in-tree HIP extension ``eventgrad_amd._core`` (gfx950).

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) so the resulting ``_core*.so`` travels with the
repo snapshot to GPU boxes. There is deliberately no JIT-compile fallback.
"""

from __future__ import annotations

import importlib

_core = None
_import_error: Exception | None = None


def _try_load():
    global _core, _import_error
    if _core is not None or _import_error is not None:
        return
    try:
        _core = importlib.import_module("eventgrad_amd._core")
    except Exception as e:  # pragma: no cover - exercised only when unbuilt
        _import_error = e


def native_available() -> bool:
    _try_load()
    return _core is not None


def native():
    """Return the extension module, raising loudly if it is not built.

    GPU code paths must call this (never guard with native_available) so a
    missing/unbuilt extension on a GPU box is an error, not a silent
    eager-PyTorch fallback.
    """
    _try_load()
    if _core is None:
        raise RuntimeError(
            "eventgrad_amd._core HIP extension is not built; run "
            "`python setup.py build_ext --inplace` (gfx950). "
            f"Original import error: {_import_error!r}"
        )
    return _core


def require_native() -> None:
    native()
