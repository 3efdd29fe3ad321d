"""Decode context parallel (DCP) all-to-all (reference parity:
flashinfer/comm/dcp_alltoall.py:227 — decode-time q/o exchange so each CP
rank attends its KV shard for every request, partials merged by LSE)."""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist

from ..parallel_attention import _merge_inplace


def dcp_scatter_q(q: torch.Tensor, group=None) -> torch.Tensor:
    r"""Replicate decode queries to every CP rank: [B, H, D] -> same on all
    ranks (allgather of the local batch shard -> full batch)."""
    world = dist.get_world_size(group)
    out = [torch.empty_like(q) for _ in range(world)]
    dist.all_gather(out, q.contiguous(), group=group)
    return torch.cat(out, dim=0)


def dcp_gather_o(
    o_partial: torch.Tensor, lse_partial: torch.Tensor, my_batch_slice: slice,
    group=None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Merge per-rank partial attention states (each rank attended its own
    KV shard for the FULL batch) and return this rank's batch slice."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    os = [torch.empty_like(o_partial) for _ in range(world)]
    ls = [torch.empty_like(lse_partial) for _ in range(world)]
    dist.all_gather(os, o_partial.contiguous(), group=group)
    dist.all_gather(ls, lse_partial.contiguous(), group=group)
    o, l = os[0].clone(), ls[0].clone().float()
    for r in range(1, world):
        _merge_inplace(o, l, os[r], ls[r].float())
    return o[my_batch_slice], l[my_batch_slice]
