"""Top-k selection (reference parity: flashinfer/topk.py top_k:522,
top_k_page_table_transform:676, top_k_ragged_transform:873 — sorting-free
threshold-search selection with the DSA sparse-attention index transforms)."""
from __future__ import annotations

from enum import IntEnum
from typing import Optional, Tuple

import torch

from ._lib import get_ext


class TopKTieBreak(IntEnum):
    NONE = 0
    PREFER_SMALLER_INDEX = 1
    PREFER_LARGER_INDEX = 2


def _sort_desc(values: torch.Tensor, indices: torch.Tensor):
    """Order (values desc, index asc as tiebreak) — used for sorted /
    deterministic output. Two stable passes: index asc, then value desc."""
    o1 = torch.argsort(indices, dim=-1, stable=True)
    v = torch.gather(values, -1, o1)
    i = torch.gather(indices, -1, o1)
    o2 = torch.argsort(v, dim=-1, descending=True, stable=True)
    return torch.gather(v, -1, o2), torch.gather(i, -1, o2)


def top_k(
    input: torch.Tensor,
    k: int,
    sorted: bool = False,
    deterministic: bool = False,
    tie_break: int = TopKTieBreak.NONE,
    dsa_graph_safe: bool = False,
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Top-k largest per row (drop-in for torch.topk at large vocab;
    unordered by default like the reference). ``sorted=True`` returns
    descending order; ``deterministic=True`` guarantees repeatable output
    ordering (implemented as the same value-desc/index-asc sort);
    ``tie_break`` selects which boundary-value indices win."""
    x2 = input.reshape(-1, input.shape[-1]).float()
    rows = x2.shape[0]
    out_v = torch.empty(rows, k, dtype=torch.float32, device=input.device)
    out_i = torch.empty(rows, k, dtype=torch.int32, device=input.device)
    get_ext().topk(x2, out_v, out_i, k, None, None, None, None, None, None,
                   1, 0, int(tie_break))
    if sorted or deterministic:
        out_v, out_i = _sort_desc(out_v, out_i)
    shape = input.shape[:-1] + (k,)
    return out_v.view(shape).to(input.dtype), out_i.view(shape).long()


def top_k_ragged_transform(
    input: torch.Tensor,
    offsets: torch.Tensor,
    lengths: torch.Tensor,
    k: int,
    deterministic: bool = False,
    tie_break: int = TopKTieBreak.NONE,
    dsa_graph_safe: bool = False,
    row_starts: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""Fused top-k + ragged index transform (DSA sparse attention stage 2):
    per row i, top-k over ``input[i, row_starts[i]:+lengths[i]]``, output
    ``local_index + offsets[i]`` (int32), padded with -1 when
    ``lengths[i] < k``."""
    x2 = input.reshape(-1, input.shape[-1]).float()
    rows = x2.shape[0]
    out_i = torch.empty(rows, k, dtype=torch.int32, device=input.device)
    get_ext().topk(
        x2, None, out_i, k, lengths.to(input.device, torch.int32),
        row_starts.to(input.device, torch.int32) if row_starts is not None else None,
        offsets.to(input.device, torch.int32), None, None, None, 1, 1,
        int(tie_break))
    return out_i


def top_k_page_table_transform(
    input: torch.Tensor,
    src_page_table: torch.Tensor,
    lengths: torch.Tensor,
    k: int,
    row_to_batch: Optional[torch.Tensor] = None,
    deterministic: bool = False,
    tie_break: int = TopKTieBreak.NONE,
    dsa_graph_safe: bool = False,
    row_starts: Optional[torch.Tensor] = None,
    page_table_row_starts: Optional[torch.Tensor] = None,
    *,
    page_size: int = 1,
    out: Optional[torch.Tensor] = None,
    out_raw_indices: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""Fused top-k + page-table index translation (DSA sparse attention):
    each selected local index ``i`` maps to
    ``src_page_table[batch, pt_start + i//page_size]*page_size + i%page_size``;
    rows shorter than k pad with -1."""
    x2 = input.reshape(-1, input.shape[-1]).float()
    rows = x2.shape[0]
    if out is None:
        out = torch.empty(rows, k, dtype=torch.int32, device=input.device)
    dev = input.device
    get_ext().topk(
        x2, None, out, k, lengths.to(dev, torch.int32),
        row_starts.to(dev, torch.int32) if row_starts is not None else None,
        None, src_page_table.to(dev, torch.int32).contiguous(),
        row_to_batch.to(dev, torch.int32) if row_to_batch is not None else None,
        page_table_row_starts.to(dev, torch.int32)
        if page_table_row_starts is not None else None,
        page_size, 2, int(tie_break))
    if out_raw_indices is not None:
        raw = top_k_ragged_transform(input, torch.zeros(rows, dtype=torch.int32,
                                                        device=dev),
                                     lengths, k, tie_break=tie_break,
                                     row_starts=row_starts)
        out_raw_indices.copy_(raw)
    return out


def top_k_varlen(
    input: torch.Tensor,        # [total_len] or [rows, max_len]
    offsets: Optional[torch.Tensor] = None,   # [rows+1] when input is flat
    lengths: Optional[torch.Tensor] = None,
    k: int = 0,
    **kwargs,
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Variable-length top-k (reference flashinfer/topk_varlen role): rows
    are ragged segments of a flat buffer (``offsets``) or padded rows with
    per-row ``lengths``. Returns (values, local indices), -1 padded."""
    if input.dim() == 1:
        if offsets is None:
            raise ValueError("flat input requires offsets")
        off = offsets.to("cpu", torch.int64)
        lens = (off[1:] - off[:-1]).to(torch.int32)
        rows = lens.numel()
        maxlen = int(lens.max()) if rows else 0
        x = torch.full((rows, max(maxlen, 1)), float("-inf"),
                       dtype=torch.float32, device=input.device)
        for r in range(rows):
            x[r, :int(lens[r])] = input[int(off[r]):int(off[r + 1])].float()
        lengths = lens.to(input.device)
    else:
        x = input.reshape(-1, input.shape[-1]).float()
        rows = x.shape[0]
        if lengths is None:
            lengths = torch.full((rows,), x.shape[1], dtype=torch.int32,
                                 device=input.device)
    out_v = torch.empty(rows, k, dtype=torch.float32, device=input.device)
    out_i = torch.empty(rows, k, dtype=torch.int32, device=input.device)
    get_ext().topk(x, out_v, out_i, k, lengths.to(input.device, torch.int32),
                   None, None, None, None, None, 1, 0, 0)
    return out_v, out_i
