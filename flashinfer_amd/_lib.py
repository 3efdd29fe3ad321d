"""Loader for the in-tree native extension ``flashinfer_amd/_C.so``.

Policy: on a machine with a GPU the native extension is REQUIRED — ops raise
immediately rather than silently falling back to eager PyTorch. On CPU-only
machines (CI containers) the Python layers (planners, wrappers, comm logic)
remain importable and testable without the .so.
"""
from __future__ import annotations

import importlib.util
from pathlib import Path

_SO = Path(__file__).resolve().parent / "_C.so"
_mod = None
_load_error: Exception | None = None


def _try_load():
    global _mod, _load_error
    if _mod is not None or _load_error is not None:
        return
    if not _SO.exists():
        _load_error = ImportError(
            f"native extension not built: {_SO} missing — run "
            "`python -m flashinfer_amd._build` (or __graft_entry__.build())"
        )
        return
    try:
        import torch  # noqa: F401  (must be loaded first for libtorch symbols)

        spec = importlib.util.spec_from_file_location("flashinfer_amd._C", _SO)
        m = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(m)
        _mod = m
    except Exception as e:  # pragma: no cover
        _load_error = e


def has_ext() -> bool:
    _try_load()
    return _mod is not None


def get_ext():
    """Return the native module; raise loudly if unavailable."""
    _try_load()
    if _mod is None:
        raise RuntimeError(
            "flashinfer_amd native extension unavailable"
        ) from _load_error
    return _mod
