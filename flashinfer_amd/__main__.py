"""CLI (reference parity: flashinfer/__main__.py — show-config:346,
module-status:595). Usage: python -m flashinfer_amd <command>."""
from __future__ import annotations

import argparse
import sys


def show_config():
    import torch

    import flashinfer_amd
    from flashinfer_amd import _lib

    print(f"flashinfer_amd {flashinfer_amd.__version__}")
    print(f"torch {torch.__version__} (hip {torch.version.hip})")
    print(f"native extension: {'loaded' if _lib.has_ext() else 'NOT BUILT'} "
          f"({_lib._SO})")
    if torch.cuda.is_available():
        p = torch.cuda.get_device_properties(0)
        print(f"device: {p.name} gcnArch={p.gcnArchName} CUs={p.multi_processor_count} "
              f"mem={p.total_memory / 2**30:.0f} GiB")
    else:
        print("device: none visible (CPU container)")
    import os

    for var in ("FLASHINFER_LOGLEVEL", "FLASHINFER_LOGDEST", "FLASHINFER_TRACE_DUMP",
                "FLASHINFER_AUTOTUNER_CACHE", "FI_OFFLOAD_ARCH"):
        if os.environ.get(var):
            print(f"{var}={os.environ[var]}")


def module_status():
    from flashinfer_amd import _lib

    if not _lib.has_ext():
        print("extension: NOT BUILT — run `python -m flashinfer_amd._build`")
        return 1
    ext = _lib.get_ext()
    ops = sorted(n for n in dir(ext) if not n.startswith("_"))
    print(f"extension: built ({_lib._SO})")
    print(f"{len(ops)} native ops:")
    for n in ops:
        print(f"  {n}")
    return 0


def build():
    from flashinfer_amd._build import build as _b

    _b(verbose=True)
    print("ok")


def main():
    ap = argparse.ArgumentParser(prog="flashinfer_amd")
    ap.add_argument("command", choices=["show-config", "module-status", "build"])
    args = ap.parse_args()
    if args.command == "show-config":
        show_config()
    elif args.command == "module-status":
        sys.exit(module_status() or 0)
    elif args.command == "build":
        build()


if __name__ == "__main__":
    main()
