"""torch symmetric-memory helpers (reference parity:
flashinfer/comm/torch_symmetric_memory.py — thin wrappers over
``torch.distributed._symmetric_memory``, which works on ROCm)."""
from __future__ import annotations

from typing import Optional

import torch


def enable_symm_mem_for_group(group_name: str) -> None:
    import torch.distributed._symmetric_memory as symm_mem

    symm_mem.enable_symm_mem_for_group(group_name)


def symm_mem_empty(*size, dtype=torch.bfloat16, device="cuda"):
    r"""Allocate a tensor from the symmetric-memory pool (rendezvous-able
    across the group's ranks)."""
    import torch.distributed._symmetric_memory as symm_mem

    return symm_mem.empty(*size, dtype=dtype, device=device)


def rendezvous(tensor: torch.Tensor, group_name: Optional[str] = None):
    import torch.distributed._symmetric_memory as symm_mem

    if group_name is None:
        import torch.distributed.distributed_c10d as c10d

        group_name = c10d._get_default_group().group_name
    return symm_mem.rendezvous(tensor, group_name)
