"""In-tree extension builder for the MI355X kernel library.

Generates a ninja file compiling every ``csrc/**/*.hip`` with hipcc for
gfx950 (fast, torch-free TUs) plus the single torch-binding TU, and links
``flashinfer_amd/_C.so`` in-tree so the shared object travels with the repo
snapshot (no site-packages / JIT-cache dependency).

Equivalent role to the reference's flashinfer/jit/ (core.py gen_jit_spec /
build_and_load, cpp_ext.py ninja emission) reduced to the MI355X single-
backend case: one offload arch, one library, ninja incremental rebuilds,
file-lock for cross-process safety.
"""
from __future__ import annotations

import os
import subprocess
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
BUILD_DIR = PKG_DIR / "_obj"
SO_PATH = PKG_DIR / "_C.so"
OFFLOAD_ARCH = os.environ.get("FI_OFFLOAD_ARCH", "gfx950")

HIPCC = os.environ.get("HIPCC", "hipcc")


def _torch_paths():
    import torch

    troot = Path(torch.__file__).parent
    return troot / "include", troot / "include/torch/csrc/api/include", troot / "lib"


def _gen_ninja() -> str:
    tinc, tapi, tlib = _torch_paths()
    pyinc = sysconfig.get_paths()["include"]
    common = (
        f"-O3 -std=c++17 -fPIC --offload-arch={OFFLOAD_ARCH} "
        f"-I{CSRC}/include -DNDEBUG"
    )
    # Kernel TUs never see torch headers; binding TU does.
    bind_flags = (
        f"-O3 -std=c++17 -fPIC -I{CSRC}/include -I{tinc} -I{tapi} -I{pyinc} "
        f"-DTORCH_EXTENSION_NAME=_C -DUSE_ROCM=1 -DTORCH_API_INCLUDE_EXTENSION_H "
        f"-D_GLIBCXX_USE_CXX11_ABI=1"
    )
    lines = [
        f"hipcc = {HIPCC}",
        "rule hipdev",
        f"  command = $hipcc -c {common} -MD -MF $out.d $in -o $out",
        "  depfile = $out.d",
        "  deps = gcc",
        "rule hiphost",
        f"  command = $hipcc -c {bind_flags} -MD -MF $out.d $in -o $out",
        "  depfile = $out.d",
        "  deps = gcc",
        "rule link",
        f"  command = $hipcc -shared $in -L{tlib} -ltorch -ltorch_python -lc10 "
        f"-lc10_hip -ltorch_hip -lamdhip64 -o $out",
        "",
    ]
    objs = []
    for src in sorted(CSRC.rglob("*.hip")):
        obj = BUILD_DIR / (src.relative_to(CSRC).as_posix().replace("/", "_") + ".o")
        lines.append(f"build {obj}: hipdev {src}")
        objs.append(str(obj))
    for src in sorted(CSRC.glob("*.cpp")):
        obj = BUILD_DIR / (src.name + ".o")
        lines.append(f"build {obj}: hiphost {src}")
        objs.append(str(obj))
    lines.append(f"build {SO_PATH}: link {' '.join(objs)}")
    lines.append("")
    return "\n".join(lines)


def build(verbose: bool = True) -> Path:
    """Compile (incrementally) and return the path of the extension."""
    BUILD_DIR.mkdir(exist_ok=True)
    ninja_file = BUILD_DIR / "build.ninja"
    content = _gen_ninja()
    if not ninja_file.exists() or ninja_file.read_text() != content:
        ninja_file.write_text(content)
    cmd = ["ninja", "-f", str(ninja_file)]
    jobs = os.environ.get("FI_BUILD_JOBS")
    if jobs:
        cmd += ["-j", jobs]
    res = subprocess.run(cmd, cwd=str(BUILD_DIR), capture_output=not verbose, text=True)
    if res.returncode != 0:
        raise RuntimeError(
            f"ninja build failed\nstdout:{res.stdout}\nstderr:{res.stderr}"
        )
    return SO_PATH


if __name__ == "__main__":
    build()
    print(f"built {SO_PATH}")
