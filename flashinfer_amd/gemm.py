"""GEMM ops on CDNA4 MFMA (reference parity: flashinfer/gemm/gemm_base.py)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace


def _as_nt(b: torch.Tensor) -> torch.Tensor:
    """Return the [N, K] K-contiguous view of b [K, N]."""
    if b.stride(0) == 1:  # column-major [K, N] — ideal
        return b.t()
    raise ValueError(
        "mm_bf16 requires the B operand in column-major ([K, N] with "
        "stride(0) == 1, e.g. `weight.t()` of a torch Linear weight). Got "
        f"strides {tuple(b.stride())}."
    )


@flashinfer_api
@fi_trace
def mm_bf16(
    a: torch.Tensor, b: torch.Tensor, out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""``out = a @ b`` on MFMA; a: [M, K] row-major, b: [K, N] column-major."""
    M, K = a.shape
    N = b.shape[1]
    if out is None:
        out = torch.empty(M, N, dtype=a.dtype, device=a.device)
    get_ext().gemm_nt(a, _as_nt(b), out, 1.0)
    return out
