"""DeepGEMM-compatible fp8 groupwise GEMM entry points (reference parity:
flashinfer/deep_gemm.py fp8_gemm_nt, m_grouped_fp8_gemm_nt_contiguous:1440,
m_grouped_fp8_gemm_nt_masked:1559 — the vendored DeepSeek GEMM API used by
MoE). On MI355X these are thin adapters over the in-house groupwise fp8 MFMA
kernel (csrc/gemm/gemm_fp8.hip): 1x128 activation scales, 128x128 weight
scales."""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .fp8_quantization import gemm_fp8_nt_groupwise, group_gemm_fp8_nt_groupwise


def _sfa_mn_major(sfa: torch.Tensor, M: int) -> torch.Tensor:
    """DeepGEMM passes sfa [M, K/128] (K-major); the kernel wants [K/128, M]."""
    if sfa.shape[0] == M:
        return sfa.t().contiguous()
    return sfa.contiguous()


def fp8_gemm_nt(
    a_fp8: Tuple[torch.Tensor, torch.Tensor],
    b_fp8: Tuple[torch.Tensor, torch.Tensor],
    d: torch.Tensor,
    c: Optional[torch.Tensor] = None,
    recipe=None,
    compiled_dims: str = "nk",
    disable_ue8m0_cast: bool = False,
) -> None:
    r"""``d = dequant(a) @ dequant(b)^T`` — a [M,K] e4m3 + sfa [M, K/128];
    b [N,K] e4m3 + sfb [K/128? N/128] blockwise."""
    a, sfa = a_fp8
    b, sfb = b_fp8
    M, K = a.shape
    N = b.shape[0]
    if sfb.shape[0] != K // 128:
        sfb = sfb.t().contiguous()
    gemm_fp8_nt_groupwise(a, b, _sfa_mn_major(sfa, M), sfb, out=d,
                          out_dtype=d.dtype)
    if c is not None:
        d += c


def m_grouped_fp8_gemm_nt_contiguous(
    a_fp8: Tuple[torch.Tensor, torch.Tensor],
    b_fp8: Tuple[torch.Tensor, torch.Tensor],
    d: torch.Tensor,
    m_indices: torch.Tensor,
    recipe=None,
    compiled_dims: str = "nk",
    disable_ue8m0_cast: bool = False,
) -> None:
    r"""Expert-contiguous grouped GEMM: row i uses expert ``m_indices[i]``
    (rows of one expert contiguous; -1 rows skipped). b [E, N, K]."""
    a, sfa = a_fp8
    b, sfb = b_fp8
    M, K = a.shape
    E, N, _ = b.shape
    # group boundaries from the sorted expert-per-row vector
    mi = m_indices.to("cpu", torch.int64)
    m_indptr = torch.searchsorted(mi, torch.arange(E + 1)).to(torch.int32)
    if sfb.dim() == 3 and sfb.shape[1] != K // 128:
        sfb = sfb.transpose(1, 2).contiguous()
    group_gemm_fp8_nt_groupwise(
        a, b, m_indptr.to(a.device), _sfa_mn_major(sfa, M), sfb, out=d,
        out_dtype=d.dtype)


def m_grouped_fp8_gemm_nt_masked(
    a_fp8: Tuple[torch.Tensor, torch.Tensor],
    b_fp8: Tuple[torch.Tensor, torch.Tensor],
    d: torch.Tensor,                         # [E, m_max, N]
    masked_m: torch.Tensor,                  # [E]
    expected_m: int,
    recipe=None,
    compiled_dims: str = "nk",
    disable_ue8m0_cast: bool = False,
) -> None:
    r"""Masked grouped GEMM: a [E, m_max, K] with the first ``masked_m[g]``
    rows of each expert valid. Valid rows are packed, run through the grouped
    kernel, and scattered back."""
    a, sfa = a_fp8
    b, sfb = b_fp8
    E, m_max, K = a.shape
    N = b.shape[1]
    mm = masked_m.to("cpu", torch.int64)
    rows = []
    for g in range(E):
        base = g * m_max
        rows.append(torch.arange(base, base + int(mm[g])))
    rows = torch.cat(rows).to(a.device) if rows else torch.empty(
        0, dtype=torch.int64, device=a.device)
    a_flat = a.reshape(E * m_max, K)[rows]
    sfa_flat = sfa.reshape(E * m_max, -1)[rows]
    m_indptr = torch.zeros(E + 1, dtype=torch.int32)
    m_indptr[1:] = mm.cumsum(0).int()
    if sfb.dim() == 3 and sfb.shape[1] != K // 128:
        sfb = sfb.transpose(1, 2).contiguous()
    packed = torch.empty(a_flat.shape[0], N, dtype=d.dtype, device=a.device)
    group_gemm_fp8_nt_groupwise(
        a_flat.contiguous(), b, m_indptr.to(a.device),
        _sfa_mn_major(sfa_flat.contiguous(), a_flat.shape[0]), sfb, out=packed,
        out_dtype=d.dtype)
    d.reshape(E * m_max, N)[rows] = packed
