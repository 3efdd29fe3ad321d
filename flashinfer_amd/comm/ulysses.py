"""Ulysses sequence-parallel attention all-to-all (reference parity:
flashinfer/comm/ulysses.py:49 UlyssesCommunicator).

scatter_heads: [B, S_local, H, D]  -> [B, S, H_local, D]   (before attention)
gather_heads:  [B, S, H_local, D]  -> [B, S_local, H, D]   (after attention)

On the 8xMI355X node this is one RCCL all_to_all_single over xGMI (every
GPU pair is a direct link, so a2a engages all 7 links at once — the right
collective shape for this fabric, unlike a ring). A gloo-compatible
allgather fallback keeps the logic CPU-testable."""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


class UlyssesCommunicator:
    def __init__(self, group: Optional[dist.ProcessGroup] = None):
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed not initialized")
        self.group = group
        self.world_size = dist.get_world_size(group)
        self.rank = dist.get_rank(group)

    def _a2a(self, x: torch.Tensor) -> torch.Tensor:
        """all-to-all over dim 0 (already arranged as [world, ...])."""
        out = torch.empty_like(x)
        backend = dist.get_backend(self.group)
        if backend == "gloo":
            # gloo lacks all_to_all_single: emulate with allgather + slice
            gathered = [torch.empty_like(x) for _ in range(self.world_size)]
            dist.all_gather(gathered, x.contiguous(), group=self.group)
            chunks = [g.chunk(self.world_size, dim=0)[self.rank] for g in gathered]
            return torch.cat(chunks, dim=0)
        dist.all_to_all_single(out, x.contiguous(), group=self.group)
        return out

    def scatter_heads(self, x: torch.Tensor) -> torch.Tensor:
        """[B, S_local, H, D] -> [B, S_local * world, H/world, D]."""
        B, S_local, H, D = x.shape
        w = self.world_size
        if H % w != 0:
            raise ValueError(f"num heads {H} not divisible by world size {w}")
        # arrange send blocks: dst rank r gets heads [r*H/w:(r+1)*H/w]
        xs = x.view(B, S_local, w, H // w, D).permute(2, 0, 1, 3, 4).contiguous()
        recv = self._a2a(xs.view(w, -1)).view(w, B, S_local, H // w, D)
        # recv[r] = rank r's seq chunk of my heads
        return recv.permute(1, 0, 2, 3, 4).reshape(B, w * S_local, H // w, D)

    def gather_heads(self, x: torch.Tensor) -> torch.Tensor:
        """[B, S, H_local, D] -> [B, S/world, H_local * world, D]."""
        B, S, Hl, D = x.shape
        w = self.world_size
        if S % w != 0:
            raise ValueError(f"seq len {S} not divisible by world size {w}")
        xs = x.view(B, w, S // w, Hl, D).permute(1, 0, 2, 3, 4).contiguous()
        recv = self._a2a(xs.view(w, -1)).view(w, B, S // w, Hl, D)
        return recv.permute(1, 2, 0, 3, 4).reshape(B, S // w, w * Hl, D)
