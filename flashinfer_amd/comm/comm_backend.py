"""Control-plane communication backend (reference parity:
flashinfer/comm/comm_backend.py). Wraps torch.distributed for object
allgather/broadcast/barrier — used for handle exchange and topology setup.
RCCL ("nccl") on GPU, gloo on CPU."""
from __future__ import annotations

from typing import Any, List, Optional

import torch
import torch.distributed as dist


class TorchDistBackend:
    def __init__(self, group: Optional[dist.ProcessGroup] = None):
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed not initialized")
        self.group = group

    @property
    def rank(self) -> int:
        return dist.get_rank(self.group)

    @property
    def world_size(self) -> int:
        return dist.get_world_size(self.group)

    def allgather_object(self, obj: Any) -> List[Any]:
        out = [None] * self.world_size
        dist.all_gather_object(out, obj, group=self.group)
        return out

    def broadcast_object(self, obj: Any, src: int = 0) -> Any:
        box = [obj if self.rank == src else None]
        dist.broadcast_object_list(box, src=src, group=self.group)
        return box[0]

    def barrier(self) -> None:
        dist.barrier(group=self.group)


def init_distributed(backend: Optional[str] = None) -> TorchDistBackend:
    """Initialize torch.distributed from env vars (torchrun-style) and return
    the default backend. Picks nccl (= RCCL on ROCm) when a GPU is visible."""
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend)
    return TorchDistBackend()
