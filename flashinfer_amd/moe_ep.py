"""MoE Expert-Parallel layer (reference parity: flashinfer/moe_ep/ — the
split-mode dispatch -> local MoE -> combine orchestration (modes/), config
enums (config.py:27-61), fault-tolerance surface (errors.py, rank masks) and
expert-placement load balancing). The reference's transports are NCCL-EP /
NIXL-EP; on an MI355X node the data plane is RCCL alltoallv over xGMI
(comm/moe_alltoall.py); mega-mode fused comm+compute kernels are a later
drop."""
from __future__ import annotations

import enum
from typing import Optional, Tuple

import torch

from .comm.moe_alltoall import MoeAlltoAll


# ---------------------------------------------------------------------------
# errors (reference moe_ep/errors.py parity)
# ---------------------------------------------------------------------------
class MoEEpNotBuiltError(RuntimeError):
    """Raised when an EP backend is invoked but its native libs are missing."""


class MoEEpFaultToleranceUnsupportedError(RuntimeError):
    """Raised when a fault-tolerance API is called but FT was not enabled."""


class MoEEpRankEvictedError(RuntimeError):
    """This rank was masked out by its peers during mask reconciliation."""


class MoEEpTransportError(RuntimeError):
    """A collective transport call failed."""

    def __init__(self, fn: str, code: int, detail: str = ""):
        self.fn, self.code, self.detail = fn, code, detail
        super().__init__(f"{fn} failed with code {code}: {detail}")


# ---------------------------------------------------------------------------
# config enums (reference moe_ep/config.py parity)
# ---------------------------------------------------------------------------
class EpAlgorithm(enum.Enum):
    SPLIT = "split"
    MEGA = "mega"


class EpLayout(enum.Enum):
    CONTIGUOUS = "contiguous"
    INTERLEAVED = "interleaved"


class QuantType(enum.Enum):
    NONE = "none"
    FP8 = "fp8"


def supports_fault_tolerance() -> bool:
    """The RCCL split transport supports the rank-mask API."""
    return True


# ---------------------------------------------------------------------------
# EPLB — expert placement load balancing
# ---------------------------------------------------------------------------
def eplb_rebalance(
    expert_load: torch.Tensor,      # [num_logical_experts] observed load
    num_ranks: int,
    num_slots_per_rank: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    r"""Greedy longest-processing-time placement with replication: experts
    sorted by load descending; each is placed on the least-loaded rank; when
    slots remain after every expert is placed once, the hottest experts get
    replicas (their load splitting across copies). Returns
    ``(phy2log [num_ranks, num_slots_per_rank], log2phy_count [E])`` — the
    physical slot map and each logical expert's replica count."""
    E = expert_load.numel()
    slots = num_ranks * num_slots_per_rank
    if slots < E:
        raise ValueError("not enough physical slots for all experts")
    load = expert_load.to("cpu", torch.float64).clone()
    replicas = torch.ones(E, dtype=torch.int64)
    # replicate hottest experts into the spare slots
    for _ in range(slots - E):
        h = int(torch.argmax(load / replicas))
        replicas[h] += 1
    # expand to physical experts with per-replica load
    phys_log = torch.repeat_interleave(torch.arange(E), replicas)
    phys_load = (load / replicas)[phys_log]
    order = torch.argsort(phys_load, descending=True)
    rank_load = torch.zeros(num_ranks, dtype=torch.float64)
    rank_fill = torch.zeros(num_ranks, dtype=torch.int64)
    phy2log = torch.full((num_ranks, num_slots_per_rank), -1, dtype=torch.int64)
    for p in order.tolist():
        open_ranks = (rank_fill < num_slots_per_rank).nonzero().flatten()
        r = int(open_ranks[torch.argmin(rank_load[open_ranks])])
        phy2log[r, rank_fill[r]] = phys_log[p]
        rank_load[r] += float(phys_load[p])
        rank_fill[r] += 1
    return phy2log, replicas


# ---------------------------------------------------------------------------
# split-mode EP layer
# ---------------------------------------------------------------------------
class MoeEp:
    r"""Split-mode EP: ``dispatch`` routes tokens to expert-owning ranks over
    RCCL alltoallv, the caller runs the local grouped MoE, ``combine``
    returns and reduces them. A rank mask (fault tolerance) removes dead
    ranks from routing: their experts' logits are masked before top-k, so
    traffic re-balances onto the survivors."""

    def __init__(self, num_experts: int, top_k: int,
                 group=None, algorithm: EpAlgorithm = EpAlgorithm.SPLIT,
                 enable_fault_tolerance: bool = False):
        import torch.distributed as dist

        self._mega = algorithm == EpAlgorithm.MEGA
        self._a2a = MoeAlltoAll(group, num_experts=num_experts, top_k=top_k)
        self._group = group
        self.num_experts = num_experts
        self.top_k = top_k
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        self.experts_per_rank = num_experts // self.world
        self._ft = enable_fault_tolerance
        self._alive = torch.ones(self.world, dtype=torch.bool)

    # ---- fault tolerance (reference FT mask API role) ----
    def mask_rank(self, rank: int) -> None:
        if not self._ft:
            raise MoEEpFaultToleranceUnsupportedError(
                "construct with enable_fault_tolerance=True")
        if rank == self.rank:
            raise MoEEpRankEvictedError("a rank cannot mask itself")
        self._alive[rank] = False

    def clear_faults(self, readmit: bool = True) -> None:
        if readmit:
            self._alive.fill_(True)

    def alive_mask(self) -> torch.Tensor:
        return self._alive.clone()

    # ---- routing + data plane ----
    def route(self, router_logits: torch.Tensor):
        """Top-k over logits with dead ranks' experts masked out."""
        logits = router_logits.float()
        if not bool(self._alive.all()):
            dead = (~self._alive).nonzero().flatten()
            for r in dead.tolist():
                logits[:, r * self.experts_per_rank:(r + 1) *
                       self.experts_per_rank] = float("-inf")
        w, ids = torch.topk(torch.softmax(logits, -1), self.top_k, dim=-1)
        w = w / w.sum(-1, keepdim=True)
        return w, ids.to(torch.int32)

    def dispatch(self, x: torch.Tensor, topk_ids: torch.Tensor):
        return self._a2a.dispatch(x, topk_ids)

    def combine(self, y: torch.Tensor, topk_weights: torch.Tensor, state):
        return self._a2a.combine(y, topk_weights, state)

    def forward_mega(self, x: torch.Tensor, router_logits: torch.Tensor,
                     expert_fn, n_chunks: int = 4) -> torch.Tensor:
        """Mega mode: CHUNKED dispatch -> expert-compute -> combine pipeline
        (reference moe_ep mega "fused comm+compute" role, MI355X-shaped:
        the a2a data plane is RCCL over xGMI, so the fusion is a software
        pipeline — chunk c+1's alltoallv is enqueued before chunk c's local
        grouped MoE runs, and RCCL's internal stream overlaps it with the
        compute. expert_fn(recv_x, recv_expert_local) -> expert_out runs the
        local experts (e.g. fused_moe's grouped GEMMs).

        Known limitation vs a true fused kernel: the count exchange inside
        dispatch synchronizes the host per chunk; the overlap won is the
        payload alltoallv (the dominant bytes), not the count handshake."""
        T = x.shape[0]
        w, ids = self.route(router_logits)
        if T == 0 or n_chunks <= 1 or T < n_chunks:
            rx, rexp, st = self._a2a.dispatch(x, ids)
            return self.combine(expert_fn(rx, rexp), w, st)
        bounds = [T * c // n_chunks for c in range(n_chunks + 1)]
        # enqueue every chunk's dispatch first: the payload a2a of later
        # chunks flows while earlier chunks compute
        pend = []
        for c in range(n_chunks):
            lo, hi = bounds[c], bounds[c + 1]
            pend.append(self._a2a.dispatch(x[lo:hi], ids[lo:hi]))
        outs = []
        for c in range(n_chunks):
            lo, hi = bounds[c], bounds[c + 1]
            rx, rexp, st = pend[c]
            y = expert_fn(rx, rexp)
            outs.append(self._a2a.combine(y, w[lo:hi], st))
        return torch.cat(outs, dim=0)

    # checkpoint hooks pass through to the data plane
    def checkpoint_prepare(self):
        return self._a2a.checkpoint_prepare()

    def checkpoint_restore(self, state):
        return self._a2a.checkpoint_restore(state)
