"""One-shot custom allreduce over hipIpc + xGMI (reference parity:
flashinfer/comm/trtllm_ar.py ONESHOT strategy + allreduce_fusion's
AR+residual+RMSNorm pattern). Each rank async-copies its input into its
hipIpc-shared buffer, then one kernel spins on the peers' Lamport sequence
flags and sums all ranks' data directly over the fabric — one kernel, no
ring, every one of the 7 xGMI links pulling concurrently.

The spin is bounded: on timeout the kernel aborts, sets a device error flag
and raises host-side — a dead peer can never hang the GPU."""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from .._lib import get_ext
from .hip_ipc import create_shared_buffer, free_shared_buffer

FLAG_BYTES = 64


class CustomAllReduce:
    r"""hipIpc one-shot allreduce workspace bound to a process group.

    Usage::

        ar = CustomAllReduce(max_bytes=16 << 20)   # collective across ranks
        y = ar.all_reduce(x)                       # sum over ranks
        out = ar.all_reduce_rmsnorm(x, residual, w)  # fused epilogue
    """

    # one-shot (each rank reads all peers' full buffers) wins while the
    # message is latency-bound; above this the two-shot RS+AG (each rank
    # pulls only its 1/world shard, then the reduced shards) wins on
    # traffic (~2x numel vs world x numel). Default from the same-die
    # probe; override with FI_AR_TWO_SHOT_BYTES, re-tune on real xGMI.
    TWO_SHOT_BYTES = 512 * 1024

    def __init__(self, max_bytes: int, group=None, device: str = "cuda",
                 spin_limit: int = 1 << 26):
        import os

        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        if self.world > 8:
            raise ValueError("one node = up to 8 ranks")
        self.max_bytes = max_bytes
        self.two_shot_bytes = int(
            os.environ.get("FI_AR_TWO_SHOT_BYTES", self.TWO_SHOT_BYTES))
        # two data slots (seq parity) + two reduced-shard slots: a rank may
        # run one call ahead of the slowest peer, so every write lands in
        # the slot of the other parity
        self._shard_bytes = max_bytes  # shard slot (worst case world=1)
        self.bufs = create_shared_buffer(
            FLAG_BYTES + 2 * max_bytes + 2 * self._shard_bytes, group)
        self.seq = 0
        self.spin_limit = spin_limit
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)
        # flags start at 0 < first seq (1) on every rank: zero the slot
        from .hip_ipc import hip_rt

        hip_rt().hipMemset(self.bufs[self.rank], 0, FLAG_BYTES)
        dist.barrier(group=self.group)

    def _push_input(self, x: torch.Tensor) -> int:
        nbytes = x.numel() * x.element_size()
        if nbytes > self.max_bytes:
            raise ValueError(f"input {nbytes}B exceeds workspace {self.max_bytes}B")
        self.seq += 1
        off = FLAG_BYTES + (self.seq % 2) * self.max_bytes
        get_ext().ipc_memcpy_to(self.bufs[self.rank] + off, x.contiguous())
        return off

    def check_errors(self):
        """On-demand timeout check (synchronizes). The per-call path never
        syncs — the kernel's bounded spin writes the device flag and aborts,
        so a dead peer corrupts one output but cannot hang the GPU; callers
        poll this at natural sync points (end of step / before a checkpoint).
        Advisor r01: a per-call .item() serialized every TP collective and
        broke hipGraph capture."""
        if int(self.error_flag.item()) != 0:
            raise RuntimeError(
                "one-shot allreduce spin timeout: a peer never arrived")

    def all_reduce(self, x: torch.Tensor,
                   out: Optional[torch.Tensor] = None,
                   strategy: Optional[str] = None) -> torch.Tensor:
        nbytes = x.numel() * x.element_size()
        off = self._push_input(x)
        if out is None:
            out = torch.empty_like(x)
        if strategy is None:
            strategy = ("two_shot" if nbytes > self.two_shot_bytes
                        and self.world > 1 else "one_shot")
        if strategy == "two_shot":
            shard_off = (FLAG_BYTES + 2 * self.max_bytes
                         + (self.seq % 2) * self._shard_bytes)
            get_ext().two_shot_all_reduce(out.view(-1), self.bufs, self.rank,
                                          self.seq, self.error_flag,
                                          self.spin_limit, off, shard_off)
        else:
            get_ext().one_shot_all_reduce(out.view(-1), self.bufs, self.rank,
                                          self.seq, self.error_flag,
                                          self.spin_limit, off)
        return out

    def all_reduce_rmsnorm(self, x: torch.Tensor, residual: Optional[torch.Tensor],
                           weight: torch.Tensor, eps: float = 1e-6,
                           out: Optional[torch.Tensor] = None) -> torch.Tensor:
        r"""Fused ``residual += allreduce(x); out = rmsnorm(residual) * w``
        (residual updated in place; pass None to norm the plain sum)."""
        off = self._push_input(x)
        if out is None:
            out = torch.empty_like(x)
        get_ext().one_shot_all_reduce_rmsnorm(
            out, residual, weight, self.bufs, self.rank, self.seq,
            self.error_flag, self.spin_limit, off, eps)
        return out

    def close(self):
        if self.bufs:
            dist.barrier(group=self.group)
            free_shared_buffer(self.bufs, self.group)
            self.bufs = None
