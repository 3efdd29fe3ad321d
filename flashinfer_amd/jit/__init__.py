"""Runtime JIT compilation of custom HIP ops (reference parity:
flashinfer/jit/core.py — JitSpec:226, gen_jit_spec:515, build_and_load with
FileLock double-check:300-321, registry:162; env paths jit/env.py).

MI355X-native: hipcc --offload-arch=gfx950 through a generated ninja file
into a content-hashed cache directory; the built library is loaded with
ctypes and exposes its ``extern "C"`` symbols. The library's device headers
(``fi/``: vec_t, MFMA fragments, online-softmax state, paged-KV descriptor,
swizzles) are on the include path — custom attention variants and fused ops
build against the same primitives the in-tree kernels use.
"""
from __future__ import annotations

import ctypes
import fcntl
import hashlib
import os
import subprocess
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional, Sequence

PKG_DIR = Path(__file__).resolve().parent.parent
FI_INCLUDE = PKG_DIR / "csrc" / "include"

_registry: Dict[str, "JitSpec"] = {}


def workspace_dir() -> Path:
    base = os.environ.get(
        "FLASHINFER_WORKSPACE_DIR",
        os.path.join(os.path.expanduser("~"), ".cache", "flashinfer_amd"),
    )
    return Path(base)


def jit_cache_dir() -> Path:
    return Path(os.environ.get("FLASHINFER_JIT_DIR", workspace_dir() / "cached_ops"))


class FileLock:
    def __init__(self, path: Path):
        self.path = path

    def __enter__(self):
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self.f = open(self.path, "w")
        fcntl.flock(self.f, fcntl.LOCK_EX)
        return self

    def __exit__(self, *a):
        fcntl.flock(self.f, fcntl.LOCK_UN)
        self.f.close()


@dataclass
class JitSpec:
    name: str
    sources: Dict[str, str]           # filename -> source text
    extra_cflags: List[str] = field(default_factory=list)
    extra_ldflags: List[str] = field(default_factory=list)
    offload_arch: str = "gfx950"
    _lib: Optional[ctypes.CDLL] = None

    @property
    def _hash(self) -> str:
        h = hashlib.sha256()
        for fn in sorted(self.sources):
            h.update(fn.encode())
            h.update(self.sources[fn].encode())
        h.update(" ".join(self.extra_cflags + self.extra_ldflags).encode())
        h.update(self.offload_arch.encode())
        return h.hexdigest()[:16]

    @property
    def build_dir(self) -> Path:
        return jit_cache_dir() / self.name / self._hash

    @property
    def so_path(self) -> Path:
        return self.build_dir / f"{self.name}.so"

    def write_ninja(self) -> Path:
        d = self.build_dir
        d.mkdir(parents=True, exist_ok=True)
        for fn, src in self.sources.items():
            p = d / fn
            if not p.exists() or p.read_text() != src:
                p.write_text(src)
        cflags = (
            f"-O3 -std=c++17 -fPIC --offload-arch={self.offload_arch} "
            f"-I{FI_INCLUDE} " + " ".join(self.extra_cflags)
        )
        lines = [
            "rule hip",
            f"  command = hipcc -c {cflags} -MD -MF $out.d $in -o $out",
            "  depfile = $out.d",
            "  deps = gcc",
            "rule link",
            f"  command = hipcc -shared $in {' '.join(self.extra_ldflags)} -o $out",
            "",
        ]
        objs = []
        for fn in sorted(self.sources):
            obj = f"{fn}.o"
            lines.append(f"build {obj}: hip {d / fn}")
            objs.append(obj)
        lines.append(f"build {self.so_path}: link {' '.join(objs)}")
        lines.append("")
        nf = d / "build.ninja"
        nf.write_text("\n".join(lines))
        return nf

    def build(self, verbose: bool = False) -> Path:
        nf = self.write_ninja()
        res = subprocess.run(
            ["ninja", "-f", str(nf)], cwd=str(self.build_dir),
            capture_output=not verbose, text=True,
        )
        if res.returncode != 0:
            raise RuntimeError(
                f"JIT build of '{self.name}' failed:\n{res.stdout}\n{res.stderr}"
            )
        return self.so_path

    def try_load(self) -> Optional[ctypes.CDLL]:
        if self.so_path.exists():
            self._lib = ctypes.CDLL(str(self.so_path))
            return self._lib
        return None

    def load(self) -> ctypes.CDLL:
        if self._lib is None:
            self._lib = ctypes.CDLL(str(self.so_path))
        return self._lib

    def build_and_load(self, verbose: bool = False) -> ctypes.CDLL:
        """Cross-process-safe double-checked build (reference jit/core.py:300)."""
        if self._lib is not None:
            return self._lib
        lib = self.try_load()
        if lib is not None:
            return lib
        with FileLock(self.build_dir.parent / f"{self._hash}.lock"):
            if not self.so_path.exists():
                self.build(verbose)
        return self.load()


def gen_jit_spec(
    name: str,
    sources: Dict[str, str],
    extra_cflags: Optional[Sequence[str]] = None,
    extra_ldflags: Optional[Sequence[str]] = None,
) -> JitSpec:
    spec = JitSpec(
        name=name, sources=dict(sources),
        extra_cflags=list(extra_cflags or []),
        extra_ldflags=list(extra_ldflags or []),
    )
    _registry[name] = spec
    return spec


def registered_specs() -> Dict[str, JitSpec]:
    return dict(_registry)
