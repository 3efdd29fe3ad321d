"""DeepSeek-V3 serving ops (reference parity: flashinfer/dsv3_ops/__init__.py
— re-exports of the narrow router GEMMs, the no-aux top-k routing and the MLA
K concat)."""
from __future__ import annotations

import torch

from .concat_ops import concat_mla_k
from .fused_moe import dsv3_routing as fused_topk_deepseek
from .gemm import mm_bf16


def mm_M1_16_K7168_N128(a: torch.Tensor, b: torch.Tensor, out=None):
    r"""DSv3 router GEMM (M in 1..16, K=7168, N=128). Skinny-M shapes route
    to hipBLASLt's latency kernels (18us class — profiles/r01_gemm_ab.txt)."""
    return mm_bf16(a, b, out=out)


def mm_M1_16_K7168_N256(a: torch.Tensor, b: torch.Tensor, out=None):
    r"""DSv3 shared-expert router GEMM variant (N=256)."""
    return mm_bf16(a, b, out=out)


__all__ = [
    "mm_M1_16_K7168_N128",
    "mm_M1_16_K7168_N256",
    "fused_topk_deepseek",
    "concat_mla_k",
]
