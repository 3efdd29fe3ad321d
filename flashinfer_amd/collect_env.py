"""Environment report (reference parity: flashinfer/collect_env.py role) —
``python -m flashinfer_amd.collect_env`` prints the ROCm / PyTorch / GPU /
framework configuration relevant to a bug report."""
from __future__ import annotations

import os
import platform
import subprocess
import sys


def _run(cmd):
    try:
        return subprocess.run(cmd, capture_output=True, text=True,
                              timeout=10).stdout.strip()
    except Exception:
        return "n/a"


def collect_env() -> dict:
    import torch

    import flashinfer_amd

    info = {
        "flashinfer_amd": getattr(flashinfer_amd, "__version__", "dev"),
        "python": sys.version.split()[0],
        "platform": platform.platform(),
        "torch": torch.__version__,
        "hip": getattr(torch.version, "hip", None),
        "cuda_available": torch.cuda.is_available(),
        "hipcc": _run(["hipcc", "--version"]).split("\n")[0] if _run(
            ["which", "hipcc"]) else "n/a",
        "PYTORCH_ROCM_ARCH": os.environ.get("PYTORCH_ROCM_ARCH", ""),
        "HSA_ENABLE_IPC_MODE_LEGACY": os.environ.get(
            "HSA_ENABLE_IPC_MODE_LEGACY", ""),
    }
    if torch.cuda.is_available():
        info["device"] = torch.cuda.get_device_name(0)
        info["gcn_arch"] = torch.cuda.get_device_properties(0).gcnArchName
        info["device_count"] = torch.cuda.device_count()
        info["rocm_smi"] = _run(["rocm-smi", "--showproductname"])[:400]
    from flashinfer_amd import _lib

    info["extension_loaded"] = _lib.has_ext()
    return info


def main():
    for k, v in collect_env().items():
        print(f"{k}: {v}")


if __name__ == "__main__":
    main()
