"""Fused gated activations (reference parity: flashinfer/activation.py)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace


def _act(which: int, input: torch.Tensor, out: Optional[torch.Tensor]) -> torch.Tensor:
    d = input.shape[-1] // 2
    if out is None:
        out = torch.empty(
            input.shape[:-1] + (d,), device=input.device, dtype=input.dtype
        )
    get_ext().act_and_mul(which, input.contiguous(), out)
    return out


@flashinfer_api
@fi_trace
def silu_and_mul(
    input: torch.Tensor, out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    r"""``silu(input[..., :d]) * input[..., d:]`` with ``d = input.shape[-1] // 2``."""
    return _act(0, input, out)


def gelu_and_mul(
    input: torch.Tensor, out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    return _act(1, input, out)


def gelu_tanh_and_mul(
    input: torch.Tensor, out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    return _act(2, input, out)
