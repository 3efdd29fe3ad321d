"""Shared helpers: layouts, dtype codes, workspace carving."""
from __future__ import annotations

import math
from typing import Optional, Tuple, Union

import torch

TensorOrPair = Union[torch.Tensor, Tuple[torch.Tensor, torch.Tensor]]

_LAYOUTS = {"NHD": 0, "HND": 1}


def layout_code(kv_layout: str) -> int:
    if kv_layout not in _LAYOUTS:
        raise ValueError(f"kv_layout must be NHD or HND, got {kv_layout}")
    return _LAYOUTS[kv_layout]


def unpack_paged_kv_cache(paged_kv_cache: TensorOrPair, kv_layout: str):
    """Accept either a single [pages, 2, ...] tensor or a (k, v) tuple of
    [pages, ...] tensors (reference contract: flashinfer/utils.py). Returns
    (k_cache, v_cache) 4-D views sharing storage."""
    if isinstance(paged_kv_cache, (tuple, list)):
        k, v = paged_kv_cache
        return k, v
    if paged_kv_cache.dim() != 5 or paged_kv_cache.size(1) != 2:
        raise ValueError(
            "single-tensor paged kv cache must be [pages, 2, ...], got "
            f"{tuple(paged_kv_cache.shape)}"
        )
    return paged_kv_cache[:, 0], paged_kv_cache[:, 1]


def check_kv_layout_dims(k_cache: torch.Tensor, kv_layout: str):
    if k_cache.dim() != 4:
        raise ValueError("paged k/v cache plane must be 4-D")


def page_size_of(k_cache: torch.Tensor, kv_layout: str) -> int:
    return k_cache.size(1) if kv_layout == "NHD" else k_cache.size(2)


def num_kv_heads_of(k_cache: torch.Tensor, kv_layout: str) -> int:
    return k_cache.size(2) if kv_layout == "NHD" else k_cache.size(1)


def default_sm_scale(head_dim: int) -> float:
    return 1.0 / math.sqrt(head_dim)


class WorkspaceAllocator:
    """Carve aligned sub-tensors out of a user-provided workspace buffer
    (equivalent role to the reference's AlignedAllocator, allocator.h)."""

    def __init__(self, buffer: torch.Tensor):
        self.buffer = buffer.view(torch.uint8).flatten()
        self.offset = 0

    def alloc(self, nbytes: int, dtype: torch.dtype, shape, align: int = 256):
        self.offset = (self.offset + align - 1) // align * align
        if self.offset + nbytes > self.buffer.numel():
            raise RuntimeError(
                f"workspace too small: need {self.offset + nbytes} bytes, "
                f"have {self.buffer.numel()} — allocate a larger workspace buffer"
            )
        t = self.buffer[self.offset : self.offset + nbytes].view(dtype).view(shape)
        self.offset += nbytes
        return t


def to_int32_device(t: torch.Tensor, device) -> torch.Tensor:
    return t.to(device=device, dtype=torch.int32, non_blocking=True)


def ceil_div(a: int, b: int) -> int:
    return (a + b - 1) // b


def round_up(a: int, b: int) -> int:
    return ceil_div(a, b) * b


def next_positive_power_of_2(x: int) -> int:
    """Smallest power of two >= max(x, 1) (reference utils role)."""
    if x < 1:
        return 1
    return 1 << (x - 1).bit_length()
