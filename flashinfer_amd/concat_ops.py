"""Concat ops (reference parity: flashinfer/concat_ops.py concat_mla_k:32)."""
from __future__ import annotations

import torch

from ._lib import get_ext


def concat_mla_k(k: torch.Tensor, k_nope: torch.Tensor,
                 k_rope: torch.Tensor) -> None:
    r"""In-place ``k[:, h] = [k_nope[:, h] | k_rope[:, 0]]`` — assembles the
    MLA K tensor by broadcasting the shared rope part to every head.
    k: [T, H, nope+rope]; k_nope: [T, H, nope]; k_rope: [T, 1, rope]."""
    get_ext().concat_mla_k(k, k_nope.contiguous(), k_rope.contiguous())
