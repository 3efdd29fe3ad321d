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
    backend: str = "auto",
) -> torch.Tensor:
    r"""``out = a @ b``; a: [M, K] row-major, b: [K, N] column-major.

    backend "auto" routes plain GEMMs to hipBLASLt (torch.mm — the library
    path for unfused GEMMs; measured on MI355X: 1564 TF at 8192^3 vs 1077 for
    the in-house kernel, and ~5-30x faster at skinny M where a 256^2 tile
    underfills the chip — profiles/r01_gemm_ab.txt). backend "mfma" forces
    the hand-written 256x256 MFMA kernel, which also backs the grouped /
    fp8-groupwise variants the library has no fused equivalent for."""
    M, K = a.shape
    N = b.shape[1]
    if backend == "auto":
        if out is None:
            return torch.mm(a, b)
        return torch.mm(a, b, out=out)
    if out is None:
        out = torch.empty(M, N, dtype=a.dtype, device=a.device)
    get_ext().gemm_nt(a, _as_nt(b), out, 1.0)
    return out


class SegmentGEMMWrapper:
    r"""Segment (grouped) GEMM for LoRA-style per-request weights (reference
    parity: flashinfer/gemm/gemm_base.py SegmentGEMMWrapper:2352). Each
    segment of x multiplies its own weight matrix (optionally indirected via
    weight_indices)."""

    def __init__(self, float_workspace_buffer: torch.Tensor, backend: str = "auto"):
        self._ws = float_workspace_buffer
        self.device = float_workspace_buffer.device

    def reset_workspace_buffer(self, float_workspace_buffer, int_workspace_buffer=None):
        self._ws = float_workspace_buffer

    def run(
        self,
        x: torch.Tensor,          # [M, K]
        weights: torch.Tensor,    # [num_w, N, K] (column-major per weight)
        batch_size: int,
        weight_column_major: bool = True,
        seg_lens: Optional[torch.Tensor] = None,
        seg_indptr: Optional[torch.Tensor] = None,
        weight_indices: Optional[torch.Tensor] = None,
        out: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if not weight_column_major:
            raise ValueError("weights must be [num_w, N, K] (column-major)")
        if seg_indptr is None:
            if seg_lens is None:
                raise ValueError("need seg_lens or seg_indptr")
            seg_indptr = torch.zeros(batch_size + 1, dtype=torch.int32,
                                     device=x.device)
            seg_indptr[1:] = seg_lens.to(x.device, torch.int64).cumsum(0).int()
        M, N = x.shape[0], weights.shape[1]
        if out is None:
            out = torch.zeros(M, N, dtype=x.dtype, device=x.device)
        max_m_tiles = (M + 127) // 128 + 1
        get_ext().group_gemm_nt(
            x, weights, out, seg_indptr.to(torch.int32),
            weight_indices.to(torch.int32) if weight_indices is not None else None,
            max_m_tiles,
        )
        return out

    forward = run


def bmm_bf16(
    a: torch.Tensor, b: torch.Tensor, out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""Batched ``out[i] = a[i] @ b[i]`` — a plain library GEMM, routed to
    hipBLASLt's strided-batched path (reference gemm_base.py bmm_bf16 role)."""
    if out is None:
        return torch.bmm(a, b)
    return torch.bmm(a, b, out=out)
