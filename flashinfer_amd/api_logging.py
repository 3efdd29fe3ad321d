"""API call logging (reference parity: flashinfer/api_logging.py
@flashinfer_api decorator:2364). Controlled by env vars:

  FLASHINFER_LOGLEVEL  0 (off, default) .. 10 (dump tensor stats)
      1: API name; 3: + shapes/dtypes of tensor args; 10: + min/max/mean
  FLASHINFER_LOGDEST   "stderr" (default), "stdout", or a path ("%i" -> pid)

Zero overhead when disabled: the decorator returns the function unchanged at
import time."""
from __future__ import annotations

import functools
import os
import sys

_LEVEL = int(os.environ.get("FLASHINFER_LOGLEVEL", "0") or "0")
_DEST = os.environ.get("FLASHINFER_LOGDEST", "stderr")


def _stream():
    if _DEST == "stderr":
        return sys.stderr
    if _DEST == "stdout":
        return sys.stdout
    path = _DEST.replace("%i", str(os.getpid()))
    return open(path, "a")


def _describe(x, level):
    import torch

    if isinstance(x, torch.Tensor):
        d = f"Tensor{tuple(x.shape)}:{str(x.dtype).replace('torch.', '')}@{x.device}"
        if level >= 10 and x.numel() > 0 and x.dtype.is_floating_point:
            xf = x.float()
            d += f"[min={xf.min().item():.4g},max={xf.max().item():.4g},mean={xf.mean().item():.4g}]"
        return d
    if isinstance(x, (list, tuple)) and len(x) <= 4:
        return type(x)(_describe(e, level) for e in x)
    return repr(x)[:80]


def flashinfer_api(fn):
    """Decorator applied to public ops; logs per FLASHINFER_LOGLEVEL."""
    if _LEVEL <= 0:
        return fn

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        s = _stream()
        if _LEVEL >= 3:
            parts = [_describe(a, _LEVEL) for a in args] + [
                f"{k}={_describe(v, _LEVEL)}" for k, v in kwargs.items()
            ]
            print(f"[flashinfer_amd] {fn.__qualname__}({', '.join(map(str, parts))})",
                  file=s)
        else:
            print(f"[flashinfer_amd] {fn.__qualname__}", file=s)
        out = fn(*args, **kwargs)
        if _LEVEL >= 10:
            print(f"[flashinfer_amd]   -> {_describe(out, _LEVEL)}", file=s)
        return out

    return wrapper
