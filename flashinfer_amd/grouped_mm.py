"""Grouped (expert-partitioned) GEMMs (reference parity:
flashinfer/grouped_mm/core.py grouped_mm_bf16:81, grouped_mm_fp8:206 —
the MoE-shaped grouped GEMM API; fp4/mxfp8 variants are N/A on CDNA4)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext
from .utils import ceil_div


def grouped_mm_bf16(
    a: torch.Tensor,          # [cum_m, k] bf16/fp16
    b: torch.Tensor,          # [E, n, k]
    m_indptr: torch.Tensor,   # [E+1] int32
    out: Optional[torch.Tensor] = None,
    out_dtype: torch.dtype = torch.bfloat16,
    *,
    backend: str = "auto",
    tactic: int = -1,
) -> torch.Tensor:
    r"""``out[s:e] = a[s:e] @ b[g]^T`` per expert g. backend "auto" uses
    hipBLASLt's grouped path (torch._grouped_mm, 1219 TF measured vs 797
    for the in-house kernel); "mfma" forces the in-house grouped kernel
    (csrc/gemm/group_gemm.hip), which also serves weight-indexed (LoRA)
    and fp8-groupwise segment GEMMs."""
    cum_m, K = a.shape
    E, N, _ = b.shape
    if (backend == "auto" and a.dtype == torch.bfloat16
            and hasattr(torch, "_grouped_mm")):
        r = torch._grouped_mm(a, b.transpose(1, 2),
                              offs=m_indptr[1:].to(a.device, torch.int32))
        if out is not None:
            out.copy_(r)
            return out
        return r
    if out is None:
        out = torch.zeros(cum_m, N, dtype=out_dtype, device=a.device)
    get_ext().group_gemm_nt(a, b, out, m_indptr.to(a.device, torch.int32),
                            None, ceil_div(cum_m, 128) + E)
    return out


def grouped_mm_fp8(
    a: torch.Tensor,          # [cum_m, k] e4m3
    b: torch.Tensor,          # [E, n, k] e4m3
    m_indptr: torch.Tensor,
    alpha: Optional[torch.Tensor] = None,   # [E] per-expert scale
    out: Optional[torch.Tensor] = None,
    out_dtype: torch.dtype = torch.bfloat16,
    **kwargs,
) -> torch.Tensor:
    r"""fp8 grouped GEMM with an optional per-expert ``alpha``. Per-expert
    scalar scaling folds into the groupwise kernel's scale tensors."""
    cum_m, K = a.shape
    E, N, _ = b.shape
    if out is None:
        out = torch.zeros(cum_m, N, dtype=out_dtype, device=a.device)
    dev = a.device
    a_scale = torch.ones(ceil_div(K, 128), cum_m, dtype=torch.float32, device=dev)
    b_scale = torch.ones(E, ceil_div(K, 128), ceil_div(N, 128),
                         dtype=torch.float32, device=dev)
    if alpha is not None:
        b_scale *= alpha.float().view(E, 1, 1)
    get_ext().gemm_fp8_grouped(
        a.view(torch.uint8), b.view(torch.uint8), out,
        m_indptr.to(dev, torch.int32), None, ceil_div(cum_m, 128) + E,
        a_scale, b_scale, 1.0, 0,
    )
    return out
