"""Sorting-free sampling ops (reference parity: flashinfer/sampling.py).

Rejection-sampling kernels take pre-drawn uniforms from the torch generator
(host side) — deterministic given a seeded generator, no in-kernel RNG state.
"""
from __future__ import annotations

from typing import Optional, Tuple, Union

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace

_ROUNDS = 32  # max rejection rounds


def _uniforms(rows: int, device, generator=None, rounds: int = _ROUNDS):
    return torch.rand(rows, rounds, device=device, generator=generator,
                      dtype=torch.float32)


def _rows(probs, indices):
    return indices.shape[0] if indices is not None else probs.shape[0]


def _prep_indices(indices):
    return indices.to(torch.int32) if indices is not None else None


def _tensor_or_scalar(x, rows, device, dtype):
    """Returns (tensor_or_None, scalar)."""
    if isinstance(x, torch.Tensor):
        return x.to(device=device, dtype=dtype).contiguous(), 0
    return None, x


def softmax(
    logits: torch.Tensor,
    temperature: Optional[Union[torch.Tensor, float]] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    r"""Online safe softmax with temperature scaling.

    Returns float32 for any input dtype (reference softmax upcasts; the
    kernel computes and writes f32 — advisor r01 fix)."""
    if isinstance(temperature, torch.Tensor):
        logits = logits.float() / temperature[:, None].clamp(min=1e-10)
        temperature = 1.0
    logits = logits.contiguous().float()
    out = torch.empty_like(logits)
    get_ext().softmax(logits, out,
                      1.0 if temperature is None else float(temperature))
    return out


def _sample(mode, from_logits, probs, indices, generator, top_k=None, top_p=None,
            scalar_p=0.0, scalar_k=0, min_p=0.0):
    probs = probs.float()
    rows = _rows(probs, indices)
    out = torch.empty(rows, dtype=torch.int32, device=probs.device)
    u = _uniforms(rows, probs.device, generator)
    tk, sk = (None, scalar_k)
    if top_k is not None:
        tk, sk = _tensor_or_scalar(top_k, rows, probs.device, torch.int32)
    tp, sp = (None, scalar_p)
    if top_p is not None:
        tp, sp = _tensor_or_scalar(top_p, rows, probs.device, torch.float32)
    get_ext().sampling(mode, from_logits, probs.contiguous(), out, u, tk, tp,
                       _prep_indices(indices), sp, sk, min_p)
    return out


@flashinfer_api
@fi_trace
def sampling_from_probs(
    probs: torch.Tensor, indices: Optional[torch.Tensor] = None,
    deterministic: bool = True, generator: Optional[torch.Generator] = None,
    check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    return _sample(0, False, probs, indices, generator)


def sampling_from_logits(
    logits: torch.Tensor, indices: Optional[torch.Tensor] = None,
    deterministic: bool = True, generator: Optional[torch.Generator] = None,
    check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    return _sample(0, True, logits, indices, generator)


def top_k_sampling_from_probs(
    probs: torch.Tensor, top_k: Union[torch.Tensor, int],
    indices: Optional[torch.Tensor] = None, deterministic: bool = True,
    generator: Optional[torch.Generator] = None, check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    return _sample(1, False, probs, indices, generator, top_k=top_k)


def top_p_sampling_from_probs(
    probs: torch.Tensor, top_p: Union[torch.Tensor, float],
    indices: Optional[torch.Tensor] = None, deterministic: bool = True,
    generator: Optional[torch.Generator] = None, check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    return _sample(2, False, probs, indices, generator, top_p=top_p)


def min_p_sampling_from_probs(
    probs: torch.Tensor, min_p: Union[torch.Tensor, float],
    indices: Optional[torch.Tensor] = None, deterministic: bool = True,
    generator: Optional[torch.Generator] = None, check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    if isinstance(min_p, torch.Tensor):
        return _sample(4, False, probs, indices, generator, top_p=min_p)
    return _sample(4, False, probs, indices, generator, min_p=float(min_p))


def top_k_top_p_sampling_from_probs(
    probs: torch.Tensor, top_k: Union[torch.Tensor, int],
    top_p: Union[torch.Tensor, float], indices: Optional[torch.Tensor] = None,
    filter_apply_order: str = "joint", deterministic: bool = True,
    generator: Optional[torch.Generator] = None, check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    if filter_apply_order == "top_k_first":
        renorm = top_k_renorm_probs(probs, top_k)
        return _sample(2, False, renorm, indices, generator, top_p=top_p)
    return _sample(3, False, probs, indices, generator, top_k=top_k, top_p=top_p)


def top_k_top_p_sampling_from_logits(
    logits: torch.Tensor, top_k: Union[torch.Tensor, int],
    top_p: Union[torch.Tensor, float], indices: Optional[torch.Tensor] = None,
    filter_apply_order: str = "joint", deterministic: bool = True,
    generator: Optional[torch.Generator] = None, check_nan: bool = False, **kwargs,
) -> torch.Tensor:
    if filter_apply_order == "top_k_first":
        masked = top_k_mask_logits(logits, top_k)
        probs = softmax(masked)
        return _sample(2, False, probs, indices, generator, top_p=top_p)
    return _sample(3, True, logits, indices, generator, top_k=top_k, top_p=top_p)


def _renorm(which, x, top_k=None, top_p=None):
    x = x.float().contiguous()
    out = torch.empty_like(x)
    tk, sk = (None, 0)
    if top_k is not None:
        tk, sk = _tensor_or_scalar(top_k, x.shape[0], x.device, torch.int32)
    tp, sp = (None, 0.0)
    if top_p is not None:
        tp, sp = _tensor_or_scalar(top_p, x.shape[0], x.device, torch.float32)
    get_ext().renorm(which, x, out, tk, tp, sp, sk)
    return out


def top_k_renorm_probs(probs: torch.Tensor, top_k: Union[torch.Tensor, int],
                       **kwargs) -> torch.Tensor:
    r"""Zero out everything below the k-th largest prob, renormalize."""
    return _renorm(0, probs, top_k=top_k)


def top_p_renorm_probs(probs: torch.Tensor, top_p: Union[torch.Tensor, float],
                       **kwargs) -> torch.Tensor:
    r"""Keep the smallest set of probs with mass >= top_p, renormalize."""
    return _renorm(1, probs, top_p=top_p)


def top_k_mask_logits(logits: torch.Tensor, top_k: Union[torch.Tensor, int],
                      **kwargs) -> torch.Tensor:
    r"""Mask logits outside the top-k to -inf."""
    return _renorm(2, logits, top_k=top_k)


def chain_speculative_sampling(
    draft_probs: torch.Tensor, draft_token_ids: torch.Tensor,
    target_probs: torch.Tensor,
    maybe_output_accepted_token_num: Optional[torch.Tensor] = None,
    maybe_output_emitted_draft_token_num: Optional[torch.Tensor] = None,
    deterministic: bool = True, generator: Optional[torch.Generator] = None,
    **kwargs,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    r"""Speculative-decoding verify-and-resample over a draft token chain."""
    B, n, V = draft_probs.shape
    dev = draft_probs.device
    out_ids = torch.empty(B, n + 1, dtype=torch.int32, device=dev)
    acc = maybe_output_accepted_token_num
    emit = maybe_output_emitted_draft_token_num
    if acc is None:
        acc = torch.zeros(B, dtype=torch.int32, device=dev)
    if emit is None:
        emit = torch.zeros(B, dtype=torch.int32, device=dev)
    u = torch.rand(B, n + 1, device=dev, generator=generator, dtype=torch.float32)
    get_ext().chain_speculative(
        draft_probs.float().contiguous(), draft_token_ids.to(torch.int32),
        target_probs.float().contiguous(), out_ids, acc, emit, u,
    )
    return out_ids, acc, emit
