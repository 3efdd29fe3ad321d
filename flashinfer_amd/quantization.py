"""Bitmask packing ops (reference parity: flashinfer/quantization/ packbits,
segment_packbits — little bitorder)."""
from __future__ import annotations

from typing import Tuple

import torch

from ._lib import get_ext
from .utils import ceil_div


def packbits(x: torch.Tensor, bitorder: str = "little") -> torch.Tensor:
    r"""Pack a boolean vector into uint8, 8 elements per byte."""
    if bitorder != "little":
        raise NotImplementedError("only little bitorder")
    xb = x.to(torch.uint8).contiguous()
    y = torch.empty(ceil_div(x.numel(), 8), dtype=torch.uint8, device=x.device)
    get_ext().packbits(xb, y)
    return y


def segment_packbits(
    x: torch.Tensor, indptr: torch.Tensor, bitorder: str = "little"
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Pack each segment x[indptr[i]:indptr[i+1]] independently; returns
    (packed, new_indptr)."""
    if bitorder != "little":
        raise NotImplementedError("only little bitorder")
    ip = indptr.to("cpu", torch.int64)
    seglens = (ip[1:] - ip[:-1]).clamp(min=0)
    out_lens = (seglens + 7) // 8
    y_indptr = torch.zeros(len(ip), dtype=torch.int32)
    y_indptr[1:] = out_lens.cumsum(0).int()
    y = torch.empty(int(y_indptr[-1]), dtype=torch.uint8, device=x.device)
    y_indptr_d = y_indptr.to(x.device)
    get_ext().segment_packbits(
        x.to(torch.uint8).contiguous(), y, indptr.to(x.device, torch.int32),
        y_indptr_d,
    )
    return y, y_indptr_d
