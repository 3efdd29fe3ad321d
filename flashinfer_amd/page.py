"""Paged KV-cache ops (reference parity: flashinfer/page.py)."""
from __future__ import annotations

from typing import Optional, Tuple, Union

import torch

from ._lib import get_ext
from .utils import layout_code, unpack_paged_kv_cache


def get_seq_lens(
    kv_indptr: torch.Tensor, kv_last_page_len: torch.Tensor, page_size: int
) -> torch.Tensor:
    r"""Per-request total KV length from the page table."""
    np_ = kv_indptr[1:] - kv_indptr[:-1]
    return torch.clamp(np_ - 1, min=0) * page_size + torch.where(
        np_ > 0, kv_last_page_len, torch.zeros_like(kv_last_page_len)
    )


def get_batch_indices_positions(
    append_indptr: torch.Tensor, seq_lens: torch.Tensor, nnz: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Convert per-request append ranges into per-token (batch_idx, position)
    pairs for :func:`append_paged_kv_cache`."""
    device = append_indptr.device
    batch_indices = torch.empty(nnz, dtype=torch.int32, device=device)
    positions = torch.empty(nnz, dtype=torch.int32, device=device)
    get_ext().get_batch_indices_positions(
        append_indptr.to(torch.int32), seq_lens.to(torch.int32), batch_indices, positions
    )
    return batch_indices, positions


def append_paged_kv_cache(
    append_key: torch.Tensor,
    append_value: torch.Tensor,
    batch_indices: torch.Tensor,
    positions: torch.Tensor,
    paged_kv_cache: Union[torch.Tensor, Tuple[torch.Tensor, torch.Tensor]],
    kv_indices: torch.Tensor,
    kv_indptr: torch.Tensor,
    kv_last_page_len: torch.Tensor,
    kv_layout: str = "NHD",
    k_scale: float = 1.0,
    v_scale: float = 1.0,
) -> None:
    r"""Scatter ``nnz`` new tokens into the paged cache.

    ``append_key``/``append_value``: [nnz, num_kv_heads, head_dim]. If the
    cache is fp8 (e4m3) and the appended values are bf16/fp16, they are
    quantized on the fly with the per-tensor scales.
    """
    k_cache, v_cache = unpack_paged_kv_cache(paged_kv_cache, kv_layout)
    get_ext().append_paged_kv_cache(
        append_key, append_value,
        batch_indices.to(torch.int32), positions.to(torch.int32),
        k_cache, v_cache,
        kv_indices.to(torch.int32), kv_indptr.to(torch.int32),
        kv_last_page_len.to(torch.int32), layout_code(kv_layout),
        k_scale, v_scale,
    )


def append_paged_mla_kv_cache(
    append_ckv: torch.Tensor,       # [nnz, 512]
    append_kpe: torch.Tensor,       # [nnz, 64]
    batch_indices: torch.Tensor,
    positions: torch.Tensor,
    ckv_cache: torch.Tensor,        # [pages, page_size, 512]
    kpe_cache: torch.Tensor,        # [pages, page_size, 64]
    kv_indices: torch.Tensor,
    kv_indptr: torch.Tensor,
    kv_last_page_len: torch.Tensor,
) -> None:
    r"""Append compressed-KV + rope-K rows into the paged MLA caches
    (reference parity: flashinfer/page.py MLA append /
    include/flashinfer/page.cuh AppendPagedKVMlaCacheKernel:870). Pure
    gather/scatter — one indexed copy per cache."""
    page_size = ckv_cache.shape[1]
    bi = batch_indices.long()
    pos = positions.long()
    page_iter = kv_indptr.long()[bi] + pos // page_size
    page_ids = kv_indices.long()[page_iter]
    entry = pos % page_size
    ckv_cache[page_ids, entry] = append_ckv.to(ckv_cache.dtype)
    kpe_cache[page_ids, entry] = append_kpe.to(kpe_cache.dtype)
