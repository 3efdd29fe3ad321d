"""FP8 (OCP e4m3fn) quantization + GEMM APIs (reference parity:
flashinfer/fp8_quantization.py mxfp8_quantize role, gemm_base.py
gemm_fp8_nt_groupwise:7589, group_gemm_fp8_nt_groupwise:8103, bmm_fp8:7387,
mm_fp8:4637)."""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ._lib import get_ext
from .utils import ceil_div

_FP8 = torch.float8_e4m3fn


def per_token_group_quant_fp8(
    x: torch.Tensor, group_size: int = 128, transpose_scale: bool = False,
    eps: float = 1e-10,
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Quantize [rows, K] to e4m3 with one f32 scale per 1x128 group.
    transpose_scale=True gives the MN-major [K/128, rows] layout the
    groupwise GEMM consumes."""
    assert group_size == 128
    rows, K = x.shape
    q = torch.empty(rows, K, dtype=torch.uint8, device=x.device)
    shape = (K // 128, rows) if transpose_scale else (rows, K // 128)
    scale = torch.empty(shape, dtype=torch.float32, device=x.device)
    get_ext().per_group_quant_fp8(x, q, scale, transpose_scale, eps)
    return q.view(_FP8), scale


def per_block_quant_mxfp8(
    w: torch.Tensor, block: int = 128
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""MX weight quantization: [128, 128] blocks with POWER-OF-TWO (e8m0)
    scales the hardware-scaled f8f6f4 MFMA applies directly — w [E, N, K] ->
    (w_fp8, scale u8 [E, K/128, N/128] e8m0 bytes). Reference mxfp8 role
    (mxfp8_quantize / mm_mxfp8) on the CDNA4-native MX path."""
    orig_shape = w.shape
    w3 = w.reshape(-1, *orig_shape[-2:])
    E, N, K = w3.shape
    wf = w3.float().view(E, N // block, block, K // block, block)
    amax = wf.abs().amax(dim=(2, 4), keepdim=True).clamp(min=1e-10)
    e = torch.ceil(torch.log2(amax / 448.0))
    scale = torch.exp2(e)
    q = (wf / scale).to(torch.float8_e4m3fn).view(E, N, K)
    sc_u8 = (e.view(E, N // block, K // block) + 127).to(torch.uint8)
    sc_u8 = sc_u8.transpose(1, 2).contiguous()  # [E, K/128, N/128]
    if len(orig_shape) == 2:
        return q.view(orig_shape), sc_u8[0]
    return q.view(orig_shape), sc_u8


def per_block_quant_fp8(
    w: torch.Tensor, block: int = 128
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Weight quantization with [128, 128] blocks: w [N, K] (or [E, N, K]) ->
    (w_fp8, scale [K/128, N/128] (or [E, K/128, N/128]))."""
    orig_shape = w.shape
    w3 = w.reshape(-1, *orig_shape[-2:])
    E, N, K = w3.shape
    wf = w3.float().view(E, N // block, block, K // block, block)
    amax = wf.abs().amax(dim=(2, 4), keepdim=True).clamp(min=1e-10)
    scale = amax / 448.0
    q = (wf / scale).clamp(-448, 448).to(_FP8).view(E, N, K)
    # -> [E, K/128, N/128]
    scale_out = scale.view(E, N // block, K // block).permute(0, 2, 1).contiguous()
    if len(orig_shape) == 2:
        return q.view(orig_shape), scale_out[0]
    return q.view(orig_shape), scale_out


def gemm_fp8_nt_groupwise(
    a: torch.Tensor, b: torch.Tensor, a_scale: torch.Tensor, b_scale: torch.Tensor,
    scale_major_mode: str = "MN", out: Optional[torch.Tensor] = None,
    out_dtype: torch.dtype = torch.bfloat16, **kwargs,
) -> torch.Tensor:
    r"""C = dequant(a) @ dequant(b)^T with 1x128 a-scales and 128x128
    b-scales. a [M, K] fp8; b [N, K] fp8; a_scale [K/128, M] (MN-major);
    b_scale [K/128, N/128]."""
    M, K = a.shape
    N = b.shape[0]
    assert scale_major_mode == "MN", "K-major scales not supported yet"
    if out is None:
        out = torch.empty(M, N, dtype=out_dtype, device=a.device)
    m_indptr = torch.tensor([0, M], dtype=torch.int32, device=a.device)
    get_ext().gemm_fp8_grouped(
        a.view(torch.uint8), b.view(torch.uint8).unsqueeze(0), out, m_indptr, None,
        ceil_div(M, 128), a_scale.contiguous(), b_scale.unsqueeze(0).contiguous(), 1.0, 0,
    )
    return out


def group_gemm_fp8_nt_groupwise(
    a: torch.Tensor, b: torch.Tensor, m_indptr: torch.Tensor, a_scale: torch.Tensor,
    b_scale: torch.Tensor, scale_major_mode: str = "MN",
    out: Optional[torch.Tensor] = None, out_dtype: torch.dtype = torch.bfloat16,
    **kwargs,
) -> torch.Tensor:
    r"""Segment/grouped variant: b [S, N, K], b_scale [S, K/128, N/128],
    segments from device m_indptr [S+1]."""
    M, K = a.shape
    N = b.shape[1]
    if out is None:
        out = torch.empty(M, N, dtype=out_dtype, device=a.device)
    get_ext().gemm_fp8_grouped(
        a.view(torch.uint8), b.view(torch.uint8), out,
        m_indptr.to(torch.int32), None, ceil_div(M, 128) + m_indptr.numel(),
        a_scale.contiguous(), b_scale.contiguous(), 1.0, 0,
    )
    return out


def bmm_fp8(
    A: torch.Tensor, B: torch.Tensor, A_scale: torch.Tensor, B_scale: torch.Tensor,
    dtype: torch.dtype = torch.bfloat16, out: Optional[torch.Tensor] = None, **kwargs,
) -> torch.Tensor:
    r"""Batched C[b] = A[b] @ B[b] * (A_scale * B_scale). A [B, M, K] fp8
    row-major; B [B, K, N] fp8 column-major (stride(1) == 1)."""
    Bb, M, K = A.shape
    N = B.shape[2]
    assert B.stride(1) == 1, "B must be column-major [B, K, N]"
    if out is None:
        out = torch.empty(Bb, M, N, dtype=dtype, device=A.device)
    m_indptr = torch.arange(0, (Bb + 1) * M, M, dtype=torch.int32, device=A.device)
    b_nt = B.transpose(1, 2)  # [B, N, K] K-contig view
    scale = float(A_scale) * float(B_scale) if not torch.is_tensor(A_scale) else float(
        A_scale.item() * B_scale.item()
    )
    get_ext().gemm_fp8_grouped(
        A.reshape(Bb * M, K).view(torch.uint8), b_nt.view(torch.uint8), out.view(Bb * M, N),
        m_indptr, None, ceil_div(M, 128), None, None, scale, 0,
    )
    return out


def mm_fp8(
    a: torch.Tensor, b: torch.Tensor, a_scale, b_scale,
    out: Optional[torch.Tensor] = None, out_dtype: torch.dtype = torch.bfloat16,
    backend: str = "auto",
    **kwargs,
) -> torch.Tensor:
    r"""Per-tensor-scaled fp8 GEMM: C = a @ b * (a_scale * b_scale);
    b column-major [K, N]. A plain library GEMM — backend "auto" uses
    hipBLASLt (torch._scaled_mm; 2.9 PF measured vs 1.2 for the in-house
    kernel, profiles/r01_gemm_ab.txt); "mfma" forces the in-house kernel
    that also backs the groupwise/batched fused variants."""
    if backend == "auto":
        sa = a_scale if torch.is_tensor(a_scale) else torch.tensor(
            float(a_scale), device=a.device)
        sb = b_scale if torch.is_tensor(b_scale) else torch.tensor(
            float(b_scale), device=a.device)
        r = torch._scaled_mm(a, b, scale_a=sa.float(), scale_b=sb.float(),
                             out_dtype=out_dtype)
        if out is not None:
            out.copy_(r)
            return out
        return r
    return bmm_fp8(a.unsqueeze(0), b.unsqueeze(0), a_scale, b_scale,
                   dtype=out_dtype, out=out.unsqueeze(0) if out is not None else None
                   ).squeeze(0)
