"""RMSNorm / LayerNorm ops (reference parity: flashinfer/norm/__init__.py)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace


def _flat2d(x: torch.Tensor) -> torch.Tensor:
    return x.reshape(-1, x.shape[-1])


@flashinfer_api
@fi_trace
def rmsnorm(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,  # accepted for API parity; no-op on MI355X
) -> torch.Tensor:
    r"""Root-mean-square normalization: ``out = input / rms(input) * weight``."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().rmsnorm(_flat2d(input), weight, _flat2d(out), eps, False)
    return out


def gemma_rmsnorm(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    r"""Gemma-style RMSNorm: ``out = input / rms(input) * (weight + 1)``."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().rmsnorm(_flat2d(input), weight, _flat2d(out), eps, True)
    return out


@flashinfer_api
@fi_trace
def fused_add_rmsnorm(
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""In-place: ``residual += input; input = rmsnorm(residual) * weight``."""
    get_ext().fused_add_rmsnorm(_flat2d(input), _flat2d(residual), weight, eps, False)


def gemma_fused_add_rmsnorm(
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    get_ext().fused_add_rmsnorm(_flat2d(input), _flat2d(residual), weight, eps, True)


def layernorm(
    input: torch.Tensor,
    gemma: torch.Tensor,
    beta: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""LayerNorm over the last dimension."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().layernorm(_flat2d(input), gemma, beta, _flat2d(out), eps)
    return out


def _scale_tensor(scale, device) -> torch.Tensor:
    if isinstance(scale, torch.Tensor):
        return scale.to(device, torch.float32)
    return torch.tensor([float(scale)], dtype=torch.float32, device=device)


@flashinfer_api
@fi_trace
def rmsnorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    weight: torch.Tensor,
    scale,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused RMSNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py rmsnorm_quant:214):
    ``out = ((input / rms(input)) * weight / scale).to(fp8_e4m3)``.
    ``scale`` is a float or a shape-(1,) f32 tensor read on device."""
    get_ext().rmsnorm_quant(_flat2d(input), None, weight, _flat2d(out),
                            _scale_tensor(scale, input.device), eps)


@flashinfer_api
@fi_trace
def fused_add_rmsnorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    scale,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused residual-add + RMSNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py fused_add_rmsnorm_quant:323). In place:
    ``residual += input; out = (rmsnorm(residual) * weight / scale).to(fp8)``."""
    get_ext().rmsnorm_quant(_flat2d(input), _flat2d(residual), weight,
                            _flat2d(out), _scale_tensor(scale, input.device), eps)


@flashinfer_api
@fi_trace
def fused_rmsnorm_silu(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    block_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""``out = silu(rmsnorm(input) * weight)`` (reference parity:
    flashinfer/norm/__init__.py fused_rmsnorm_silu:689; the fp4 block_scale
    output is N/A on CDNA4)."""
    if block_scale is not None:
        raise NotImplementedError("fp4 block scales are N/A on CDNA4")
    if out is None:
        out = torch.empty_like(input)
    assert out.dtype == input.dtype, "out dtype must match input on CDNA4 path"
    get_ext().rmsnorm_silu(_flat2d(input), weight, _flat2d(out), eps)
    return out


@flashinfer_api
@fi_trace
def layernorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    weight: torch.Tensor,
    scale,
    bias: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused LayerNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py layernorm_quant role):
    ``out = (((input - mean) / sqrt(var + eps)) * weight (+ bias)
    / scale).to(fp8_e4m3)``."""
    get_ext().layernorm_quant(_flat2d(input), weight, bias, _flat2d(out),
                              _scale_tensor(scale, input.device), eps)
