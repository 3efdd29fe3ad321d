"""Call-signature tracing (reference parity: flashinfer/fi_trace.py —
per-API definition JSON dumps for flashinfer-bench-style replay).
Enable with FLASHINFER_TRACE_DUMP=<dir>; each traced call appends one JSON
line per unique signature to <dir>/<api>.jsonl."""
from __future__ import annotations

import functools
import json
import os
from pathlib import Path

_DIR = os.environ.get("FLASHINFER_TRACE_DUMP", "")
_seen = set()


def _sig(x):
    import torch

    if isinstance(x, torch.Tensor):
        return {"shape": list(x.shape), "dtype": str(x.dtype).replace("torch.", "")}
    if isinstance(x, (int, float, bool, str, type(None))):
        return x
    if isinstance(x, (list, tuple)):
        return [_sig(e) for e in x]
    return repr(type(x))


def fi_trace(fn):
    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        # trace_apply hook: a registered substitute takes over this call
        from . import trace_apply as _ta

        sub = _ta._dispatch(fn.__name__) if _ta.is_enabled() else None
        if sub is not None:
            return sub(*args, **kwargs)
        if _DIR:
            rec = {
                "api": fn.__qualname__,
                "args": [_sig(a) for a in args],
                "kwargs": {k: _sig(v) for k, v in kwargs.items()},
            }
            key = json.dumps(rec, sort_keys=True, default=str)
            if key not in _seen:
                _seen.add(key)
                Path(_DIR).mkdir(parents=True, exist_ok=True)
                with open(Path(_DIR) / f"{fn.__name__}.jsonl", "a") as f:
                    f.write(key + "\n")
        return fn(*args, **kwargs)

    return wrapper
