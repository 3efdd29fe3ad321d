"""Runtime kernel substitution for fi_trace definitions (reference parity:
flashinfer/trace_apply/__init__.py — enable_apply/disable_apply/is_enabled/
stats). ``enable_apply({api_name: callable})`` reroutes any
``@fi_trace``-decorated API through the registered callable; env hook:
``FLASHINFER_TRACE_APPLY=1`` + ``FLASHINFER_TRACE_APPLY_PATH=mod:attr``
(a module attribute holding the mapping)."""
from __future__ import annotations

import importlib
import os
from collections import Counter
from typing import Callable, Dict, Optional

_registry: Dict[str, Callable] = {}
_enabled = False
_stats: Counter = Counter()


def enable_apply(mapping: Dict[str, Callable]) -> None:
    """Register substitutes keyed by the fi_trace api name (the decorated
    function's __name__) and turn dispatch on."""
    global _enabled
    _registry.update(mapping)
    _enabled = True


def disable_apply() -> None:
    global _enabled
    _enabled = False
    _registry.clear()
    _stats.clear()


def is_enabled() -> bool:
    return _enabled


def stats() -> dict:
    """Per-(api, status) dispatch counts: hit / fallback_no_candidate."""
    return dict(_stats)


stats_snapshot = stats


def _dispatch(name: str) -> Optional[Callable]:
    """Called by the fi_trace wrapper on every invocation."""
    if not _enabled:
        return None
    fn = _registry.get(name)
    _stats[(name, "hit" if fn else "fallback_no_candidate")] += 1
    return fn


def _enable_apply_from_env() -> None:
    if os.environ.get("FLASHINFER_TRACE_APPLY", "") != "1":
        return
    path = os.environ.get("FLASHINFER_TRACE_APPLY_PATH", "")
    if not path or ":" not in path:
        return
    mod, attr = path.split(":", 1)
    mapping = getattr(importlib.import_module(mod), attr)
    enable_apply(dict(mapping))
