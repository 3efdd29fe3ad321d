"""Rotary positional embedding ops (reference parity: flashinfer/rope.py).

All variants funnel into one kernel launch rotating q and k together, driven
by per-token ``pos_ids``:
  * apply_rope(_inplace): ragged batch via (indptr, offsets) — pos_ids are
    derived on device.
  * apply_rope_pos_ids(_inplace): explicit positions.
  * apply_llama31_rope(_pos_ids)(_inplace): llama-3.1 wavelength-scaled freqs.
  * apply_rope_with_cos_sin_cache(_inplace): vLLM/SGLang-style f32 cache
    [max_pos, rot_dim] = [cos | sin] — the fast serving path (host-precomputed
    trig; on-device sin/cos turns this memory-bound op VALU-bound on CDNA4).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from ._lib import get_ext


def _pos_ids_from_indptr(indptr: torch.Tensor, offsets: torch.Tensor, nnz: int):
    # pos_ids[i] = offsets[b] + (i - indptr[b]) for i in segment b
    indptr = indptr.to(torch.int64)
    lens = indptr[1:] - indptr[:-1]
    batch_ids = torch.repeat_interleave(
        torch.arange(len(lens), device=indptr.device), lens
    )
    local = torch.arange(nnz, device=indptr.device) - indptr[batch_ids]
    return (offsets.to(torch.int64)[batch_ids] + local).to(torch.int32)


def _run_rope(
    q, k, q_out, k_out, pos_ids,
    cos_sin_cache=None, rotary_dim=None, interleave=False,
    rope_scale=1.0, rope_theta=1e4,
    low_freq_factor=None, high_freq_factor=None, old_context_len=None,
):
    head_dim = q.shape[-1]
    rot_dim = rotary_dim if rotary_dim is not None else (
        cos_sin_cache.shape[-1] if cos_sin_cache is not None else head_dim
    )
    smooth_a = smooth_b = 0.0
    rcp_factor = 1.0
    if low_freq_factor is not None:
        # llama-3.1: smooth interpolation between freq/factor and freq by
        # wavelength band (see reference pos_enc.cuh:1528)
        smooth_a = old_context_len / (2 * math.pi) / (high_freq_factor - low_freq_factor)
        smooth_b = -low_freq_factor / (high_freq_factor - low_freq_factor)
        rcp_factor = 1.0 / rope_scale
        rope_scale = 1.0  # scaling handled via rcp_factor path
    get_ext().apply_rope(
        q, k, q_out, k_out, pos_ids.to(torch.int32),
        cos_sin_cache, rot_dim, interleave, rope_scale, rope_theta,
        smooth_a, smooth_b, rcp_factor,
    )
    return q_out, k_out


def apply_rope(
    q: torch.Tensor, k: torch.Tensor, indptr: torch.Tensor, offsets: torch.Tensor,
    rotary_dim: Optional[int] = None, interleave: bool = False,
    rope_scale: float = 1, rope_theta: float = 1e4,
) -> Tuple[torch.Tensor, torch.Tensor]:
    pos_ids = _pos_ids_from_indptr(indptr, offsets, q.shape[0])
    return _run_rope(q, k, torch.empty_like(q), torch.empty_like(k), pos_ids,
                     rotary_dim=rotary_dim, interleave=interleave,
                     rope_scale=rope_scale, rope_theta=rope_theta)


def apply_rope_inplace(
    q, k, indptr, offsets, rotary_dim=None, interleave=False,
    rope_scale: float = 1, rope_theta: float = 1e4,
) -> None:
    pos_ids = _pos_ids_from_indptr(indptr, offsets, q.shape[0])
    _run_rope(q, k, q, k, pos_ids, rotary_dim=rotary_dim, interleave=interleave,
              rope_scale=rope_scale, rope_theta=rope_theta)


def apply_rope_pos_ids(
    q, k, pos_ids, rotary_dim=None, interleave=False,
    rope_scale: float = 1, rope_theta: float = 1e4,
) -> Tuple[torch.Tensor, torch.Tensor]:
    return _run_rope(q, k, torch.empty_like(q), torch.empty_like(k), pos_ids,
                     rotary_dim=rotary_dim, interleave=interleave,
                     rope_scale=rope_scale, rope_theta=rope_theta)


def apply_rope_pos_ids_inplace(
    q, k, pos_ids, rotary_dim=None, interleave=False,
    rope_scale: float = 1, rope_theta: float = 1e4,
) -> None:
    _run_rope(q, k, q, k, pos_ids, rotary_dim=rotary_dim, interleave=interleave,
              rope_scale=rope_scale, rope_theta=rope_theta)


def apply_llama31_rope(
    q, k, indptr, offsets, rotary_dim=None, interleave=False,
    rope_scale: float = 8, rope_theta: float = 5e5,
    low_freq_factor: float = 1, high_freq_factor: float = 4,
    old_context_len: int = 8192,
) -> Tuple[torch.Tensor, torch.Tensor]:
    pos_ids = _pos_ids_from_indptr(indptr, offsets, q.shape[0])
    return _run_rope(q, k, torch.empty_like(q), torch.empty_like(k), pos_ids,
                     rotary_dim=rotary_dim, interleave=interleave,
                     rope_scale=rope_scale, rope_theta=rope_theta,
                     low_freq_factor=low_freq_factor,
                     high_freq_factor=high_freq_factor,
                     old_context_len=old_context_len)


def apply_llama31_rope_inplace(
    q, k, indptr, offsets, rotary_dim=None, interleave=False,
    rope_scale: float = 8, rope_theta: float = 5e5,
    low_freq_factor: float = 1, high_freq_factor: float = 4,
    old_context_len: int = 8192,
) -> None:
    pos_ids = _pos_ids_from_indptr(indptr, offsets, q.shape[0])
    _run_rope(q, k, q, k, pos_ids, rotary_dim=rotary_dim, interleave=interleave,
              rope_scale=rope_scale, rope_theta=rope_theta,
              low_freq_factor=low_freq_factor, high_freq_factor=high_freq_factor,
              old_context_len=old_context_len)


def apply_llama31_rope_pos_ids(
    q, k, pos_ids, rotary_dim=None, interleave=False,
    rope_scale: float = 8, rope_theta: float = 5e5,
    low_freq_factor: float = 1, high_freq_factor: float = 4,
    old_context_len: int = 8192,
) -> Tuple[torch.Tensor, torch.Tensor]:
    return _run_rope(q, k, torch.empty_like(q), torch.empty_like(k), pos_ids,
                     rotary_dim=rotary_dim, interleave=interleave,
                     rope_scale=rope_scale, rope_theta=rope_theta,
                     low_freq_factor=low_freq_factor,
                     high_freq_factor=high_freq_factor,
                     old_context_len=old_context_len)


def apply_llama31_rope_pos_ids_inplace(
    q, k, pos_ids, rotary_dim=None, interleave=False,
    rope_scale: float = 8, rope_theta: float = 5e5,
    low_freq_factor: float = 1, high_freq_factor: float = 4,
    old_context_len: int = 8192,
) -> None:
    _run_rope(q, k, q, k, pos_ids, rotary_dim=rotary_dim, interleave=interleave,
              rope_scale=rope_scale, rope_theta=rope_theta,
              low_freq_factor=low_freq_factor, high_freq_factor=high_freq_factor,
              old_context_len=old_context_len)


def _csc_reshape(x: torch.Tensor, head_size: int):
    # vLLM convention: [..., num_heads*head_size] flat; reshape to 3-D.
    # Guard: a 3-D [nnz, H, D] input here would silently become nnz*H
    # "tokens" and index pos_ids out of bounds (native crash).
    if x.dim() == 3 and x.shape[-1] == head_size:
        return x
    if x.shape[-1] % head_size:
        raise ValueError(
            f"last dim {x.shape[-1]} is not a multiple of head_size "
            f"{head_size} (expected flat [tokens, num_heads*head_size])")
    nnz = x.numel() // x.shape[-1]
    return x.view(nnz, x.shape[-1] // head_size, head_size)


def apply_rope_with_cos_sin_cache(
    positions: torch.Tensor, query: torch.Tensor, key: torch.Tensor,
    head_size: int, cos_sin_cache: torch.Tensor, is_neox: bool = True,
) -> Tuple[torch.Tensor, torch.Tensor]:
    q_out = torch.empty_like(query)
    k_out = torch.empty_like(key)
    _run_rope(
        _csc_reshape(query, head_size), _csc_reshape(key, head_size),
        _csc_reshape(q_out, head_size), _csc_reshape(k_out, head_size),
        positions, cos_sin_cache=cos_sin_cache.float(), interleave=not is_neox,
    )
    return q_out, k_out


def apply_rope_quantize_append(
    q, k, v, pos_ids, paged_kv_cache, kv_indices, kv_indptr, kv_last_page_len,
    batch_indices, positions, cos_sin_cache=None, rotary_dim=None,
    interleave=False, rope_theta: float = 1e4, k_scale: float = 1.0,
    v_scale: float = 1.0, kv_layout: str = "NHD",
):
    r"""RoPE -> (fp8) quantize -> paged append (reference
    RopeQuantizeAppendPagedKVCacheKernel pos_enc.cuh:808 role; composed from
    the rope and quantizing-append kernels here — two launches)."""
    from .page import append_paged_kv_cache

    q_out, k_out = _run_rope(
        q, k, torch.empty_like(q), torch.empty_like(k), pos_ids,
        cos_sin_cache=cos_sin_cache, rotary_dim=rotary_dim,
        interleave=interleave, rope_theta=rope_theta,
    )
    append_paged_kv_cache(k_out, v, batch_indices, positions, paged_kv_cache,
                          kv_indices, kv_indptr, kv_last_page_len, kv_layout,
                          k_scale=k_scale, v_scale=v_scale)
    return q_out


def apply_rope_with_cos_sin_cache_inplace(
    positions: torch.Tensor, query: torch.Tensor, key: torch.Tensor,
    head_size: int, cos_sin_cache: torch.Tensor, is_neox: bool = True,
) -> None:
    q3 = _csc_reshape(query, head_size)
    k3 = _csc_reshape(key, head_size)
    _run_rope(q3, k3, q3, k3, positions,
              cos_sin_cache=cos_sin_cache.float(), interleave=not is_neox)
