"""Kernel-tactic autotuner (reference parity: flashinfer/autotuner/
autotuner.py — autotune() context manager:783, TunableRunner:695,
TuningConfig:435, JSON cache keyed on GPU + library version).

MI355X design: one backend per op, so tactics are tile/schedule variants of
the same HIP kernel (e.g. GEMM v1 128^2 vs v2 256^2 pipeline, decode chunk
targets). Profiling runs under the autotune() context; chosen tactics are
cached in-memory and optionally persisted to JSON."""
from __future__ import annotations

import contextlib
import json
import os
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Tuple

_tuning_enabled = False
_cache: Dict[str, int] = {}


def _cache_path() -> Path:
    base = os.environ.get(
        "FLASHINFER_AUTOTUNER_CACHE",
        os.path.join(os.path.expanduser("~"), ".cache", "flashinfer_amd",
                     "tuning_cache.json"),
    )
    return Path(base)


def load_cache() -> None:
    p = _cache_path()
    if p.exists():
        try:
            _cache.update(json.loads(p.read_text()))
        except Exception:
            pass


def save_cache() -> None:
    p = _cache_path()
    p.parent.mkdir(parents=True, exist_ok=True)
    p.write_text(json.dumps(_cache, indent=1, sort_keys=True))


@contextlib.contextmanager
def autotune(enable: bool = True, persist: bool = False):
    r"""Within this context, TunableRunner.run profiles all tactics on first
    sight of each (op, bucketed-shape) key and records the winner; outside
    it, the recorded (or default) tactic runs directly."""
    global _tuning_enabled
    prev = _tuning_enabled
    _tuning_enabled = enable
    try:
        yield
    finally:
        _tuning_enabled = prev
        if persist:
            save_cache()


def _bucket(n: int) -> int:
    # power-of-two shape bucketing (reference bucketed dynamic dims :98-190)
    b = 1
    while b < n:
        b <<= 1
    return b


@dataclass
class TuningConfig:
    name: str
    dynamic_dims: Tuple[int, ...] = ()


class TunableRunner:
    r"""Wraps a list of tactic callables (same signature); selects by
    profiling under autotune()."""

    def __init__(self, name: str, tactics: List[Callable], key_fn=None):
        self.name = name
        self.tactics = tactics
        self.key_fn = key_fn or (lambda *a, **k: ())

    def _key(self, *args, **kwargs) -> str:
        dims = tuple(_bucket(int(d)) for d in self.key_fn(*args, **kwargs))
        return f"{self.name}:{dims}"

    def run(self, *args, **kwargs):
        import torch

        key = self._key(*args, **kwargs)
        if key in _cache:
            return self.tactics[_cache[key] % len(self.tactics)](*args, **kwargs)
        if not _tuning_enabled or len(self.tactics) == 1:
            return self.tactics[0](*args, **kwargs)
        best, best_t = 0, float("inf")
        out = None
        for i, t in enumerate(self.tactics):
            try:
                t(*args, **kwargs)  # warm
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(3):
                    out = t(*args, **kwargs)
                torch.cuda.synchronize()
                dt = time.perf_counter() - t0
            except Exception:
                continue
            if dt < best_t:
                best, best_t = i, dt
        _cache[key] = best
        return out if out is not None else self.tactics[best](*args, **kwargs)


load_cache()
