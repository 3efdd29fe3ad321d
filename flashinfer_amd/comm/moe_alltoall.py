"""MoE expert-parallel dispatch/combine (reference parity:
flashinfer/comm/trtllm_moe_alltoall.py MoeAlltoAll:790 — the throughput a2a).

MI355X design: variable all-to-all (alltoallv) over RCCL/xGMI. Dispatch
routes each token's top-k expert copies to the rank owning that expert;
combine returns the expert outputs and reduces them with the routing
weights. Count exchange is a small fixed all_to_all; payloads go through
``dist.all_to_all_single`` with split sizes. A gloo fallback (allgather)
keeps the routing logic CPU-testable.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist


def _a2a_v(x: torch.Tensor, out_splits, in_splits, group):
    backend = dist.get_backend(group)
    total_out = int(sum(out_splits))
    out = torch.empty(total_out, *x.shape[1:], dtype=x.dtype, device=x.device)
    if backend == "gloo":
        w = dist.get_world_size(group)
        rank = dist.get_rank(group)
        # emulate with all_gather_object (CPU tests only)
        sent = list(x.split(in_splits, dim=0))
        boxes = [None] * w
        dist.all_gather_object(boxes, [t.cpu() for t in sent], group=group)
        parts = [boxes[r][rank].to(x.device) for r in range(w)]
        torch.cat(parts, dim=0, out=out)
        return out
    dist.all_to_all_single(out, x.contiguous(), out_splits, in_splits, group=group)
    return out


class MoeAlltoAll:
    r"""Expert-parallel token dispatch/combine.

    num_experts must be divisible by the EP world size; expert e lives on
    rank ``e // (num_experts // world)``.
    """

    def __init__(self, group: Optional[dist.ProcessGroup] = None,
                 num_experts: int = 8, top_k: int = 2, max_tokens: int = 0):
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed not initialized")
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        if num_experts % self.world != 0:
            raise ValueError("num_experts must divide evenly across EP ranks")
        self.num_experts = num_experts
        self.experts_per_rank = num_experts // self.world
        self.top_k = top_k

    def dispatch(
        self, x: torch.Tensor, topk_ids: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, dict]:
        """Route token copies to expert-owner ranks.

        x: [T, hidden]; topk_ids: [T, top_k] global expert ids.
        Returns (recv_x [R, hidden], recv_expert_local [R], state for combine).
        """
        T, K = topk_ids.shape
        flat_exp = topk_ids.reshape(-1)  # [T*K]
        dest = flat_exp // self.experts_per_rank  # owner rank per copy
        order = torch.argsort(dest, stable=True)  # group copies by dest rank
        send_x = x.repeat_interleave(K, dim=0)[order]
        send_exp = flat_exp[order]
        in_splits = torch.bincount(dest, minlength=self.world)
        # exchange counts
        out_splits = torch.empty_like(in_splits)
        if dist.get_backend(self.group) == "gloo":
            boxes = [None] * self.world
            dist.all_gather_object(boxes, in_splits.tolist(), group=self.group)
            out_splits = torch.tensor([boxes[r][self.rank] for r in range(self.world)],
                                      dtype=in_splits.dtype, device=in_splits.device)
        else:
            dist.all_to_all_single(out_splits, in_splits.contiguous(),
                                   group=self.group)
        in_l = in_splits.tolist()
        out_l = out_splits.tolist()
        recv_x = _a2a_v(send_x, out_l, in_l, self.group)
        recv_exp = _a2a_v(send_exp.unsqueeze(1), out_l, in_l, self.group).squeeze(1)
        state = dict(order=order, in_splits=in_l, out_splits=out_l, T=T, K=K,
                     hidden=x.shape[-1])
        return recv_x, recv_exp - self.rank * self.experts_per_rank, state

    def combine(
        self, expert_out: torch.Tensor, topk_weights: torch.Tensor, state: dict
    ) -> torch.Tensor:
        """Return expert outputs to source ranks and reduce with weights.

        expert_out: [R, hidden] aligned with dispatch's recv_x.
        topk_weights: [T, top_k].
        """
        back = _a2a_v(expert_out, state["in_splits"], state["out_splits"], self.group)
        T, K = state["T"], state["K"]
        if T == 0:
            # zero-token idle rank (reference repro_ikr_zero_token_idle
            # contract): the alltoallv above MUST still run — peers'
            # splits reference this rank — but there is nothing to reduce
            # locally and view(0, K, -1) is ambiguous.
            return expert_out.new_zeros((0, state["hidden"]))
        inv = torch.empty_like(state["order"])
        inv[state["order"]] = torch.arange(len(state["order"]),
                                           device=state["order"].device)
        contrib = back[inv].view(T, K, -1)
        return (contrib * topk_weights.unsqueeze(-1).to(contrib.dtype)).sum(dim=1)

    # ---- workspace checkpoint/restore (reference trtllm_moe_alltoall.py
    # :1052/:1072 contract — engines snapshot comm state across hipGraph
    # re-capture; the RCCL data plane is stateless, so this captures only the
    # config) ----
    def checkpoint_prepare(self) -> dict:
        return dict(num_experts=self.num_experts, top_k=self.top_k,
                    world=self.world, rank=self.rank)

    def checkpoint_restore(self, state: dict) -> None:
        assert state["num_experts"] == self.num_experts
        assert state["world"] == self.world
