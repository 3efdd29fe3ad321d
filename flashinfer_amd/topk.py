"""Top-k selection (reference parity: flashinfer/topk.py top_k — unordered
largest-k values+indices via sorting-free threshold search)."""
from __future__ import annotations

from typing import Tuple

import torch

from ._lib import get_ext


def top_k(x: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Unordered top-k along the last dim. x [rows, d] -> (values [rows, k],
    indices [rows, k])."""
    x2 = x.reshape(-1, x.shape[-1]).float()
    rows = x2.shape[0]
    out_v = torch.empty(rows, k, dtype=torch.float32, device=x.device)
    out_i = torch.empty(rows, k, dtype=torch.int32, device=x.device)
    get_ext().topk(x2, out_v, out_i, k)
    shape = x.shape[:-1] + (k,)
    return out_v.view(shape).to(x.dtype), out_i.view(shape)
