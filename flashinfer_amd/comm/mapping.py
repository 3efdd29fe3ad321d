"""Parallel topology mapping (reference parity: flashinfer/comm/mapping.py:21).

Decomposes a world of ranks into tp / pp / cp and MoE (moe_tp x moe_ep)
sub-groups, with optional attention-dp. Rank layout (innermost last):
``rank = pp_rank * (cp_size * tp_size) + cp_rank * tp_size + tp_rank``.
Pure Python — used to build torch.distributed (RCCL) process groups on the
8xMI355X node where every GPU pair is one xGMI hop.
"""
from __future__ import annotations

from typing import List, Optional


class Mapping:
    def __init__(
        self,
        world_size: int = 1,
        rank: int = 0,
        gpus_per_node: int = 8,
        tp_size: int = 1,
        pp_size: int = 1,
        cp_size: int = 1,
        moe_tp_size: int = -1,
        moe_ep_size: int = -1,
        attn_tp_size: int = -1,
        attn_cp_size: int = -1,
        enable_attention_dp: bool = False,
    ):
        if tp_size * pp_size * cp_size != world_size:
            raise ValueError(
                f"world_size ({world_size}) != tp ({tp_size}) * pp ({pp_size})"
                f" * cp ({cp_size})"
            )
        # MoE defaults: moe_tp x moe_ep covers the tp*cp slice
        if moe_tp_size == -1 and moe_ep_size == -1:
            moe_tp_size, moe_ep_size = tp_size * cp_size, 1
        elif moe_tp_size == -1:
            moe_tp_size = tp_size * cp_size // moe_ep_size
        elif moe_ep_size == -1:
            moe_ep_size = tp_size * cp_size // moe_tp_size
        if moe_tp_size * moe_ep_size != tp_size * cp_size:
            raise ValueError("moe_tp_size * moe_ep_size must equal tp_size * cp_size")
        # attention tp/cp split (Ulysses): defaults to the dense split
        if attn_tp_size == -1 and attn_cp_size == -1:
            attn_tp_size, attn_cp_size = tp_size, cp_size
        elif attn_tp_size == -1:
            attn_tp_size = tp_size * cp_size // attn_cp_size
        elif attn_cp_size == -1:
            attn_cp_size = tp_size * cp_size // attn_tp_size
        if attn_tp_size * attn_cp_size != tp_size * cp_size:
            raise ValueError("attn_tp * attn_cp must equal tp * cp")

        self.world_size = world_size
        self.rank = rank
        self.gpus_per_node = gpus_per_node
        self.tp_size = tp_size
        self.pp_size = pp_size
        self.cp_size = cp_size
        self.moe_tp_size = moe_tp_size
        self.moe_ep_size = moe_ep_size
        self.attn_tp_size = attn_tp_size
        self.attn_cp_size = attn_cp_size
        self.enable_attention_dp = enable_attention_dp

    # ---- rank coordinates ----
    @property
    def pp_rank(self) -> int:
        return self.rank // (self.cp_size * self.tp_size)

    @property
    def cp_rank(self) -> int:
        return (self.rank % (self.cp_size * self.tp_size)) // self.tp_size

    @property
    def tp_rank(self) -> int:
        return self.rank % self.tp_size

    @property
    def tp_cp_rank(self) -> int:
        return self.rank % (self.cp_size * self.tp_size)

    @property
    def moe_tp_rank(self) -> int:
        return self.tp_cp_rank // self.moe_ep_size

    @property
    def moe_ep_rank(self) -> int:
        return self.tp_cp_rank % self.moe_ep_size

    @property
    def attn_tp_rank(self) -> int:
        return self.tp_cp_rank % self.attn_tp_size

    @property
    def attn_cp_rank(self) -> int:
        return self.tp_cp_rank // self.attn_tp_size

    @property
    def node_rank(self) -> int:
        return self.rank // self.gpus_per_node

    @property
    def local_rank(self) -> int:
        return self.rank % self.gpus_per_node

    # ---- groups (lists of global ranks) ----
    def _pp_base(self) -> int:
        return self.pp_rank * self.cp_size * self.tp_size

    @property
    def tp_group(self) -> List[int]:
        base = self._pp_base() + self.cp_rank * self.tp_size
        return list(range(base, base + self.tp_size))

    @property
    def cp_group(self) -> List[int]:
        base = self._pp_base() + self.tp_rank
        return [base + i * self.tp_size for i in range(self.cp_size)]

    @property
    def pp_group(self) -> List[int]:
        off = self.rank % (self.cp_size * self.tp_size)
        return [off + i * self.cp_size * self.tp_size for i in range(self.pp_size)]

    @property
    def moe_ep_group(self) -> List[int]:
        base = self._pp_base() + self.moe_tp_rank * self.moe_ep_size
        return list(range(base, base + self.moe_ep_size))

    @property
    def moe_tp_group(self) -> List[int]:
        base = self._pp_base() + self.moe_ep_rank
        return [base + i * self.moe_ep_size for i in range(self.moe_tp_size)]

    @property
    def attn_tp_group(self) -> List[int]:
        base = self._pp_base() + self.attn_cp_rank * self.attn_tp_size
        return list(range(base, base + self.attn_tp_size))

    @property
    def attn_cp_group(self) -> List[int]:
        base = self._pp_base() + self.attn_tp_rank
        return [base + i * self.attn_tp_size for i in range(self.attn_cp_size)]

    # ---- convenience ----
    def is_first_pp_rank(self) -> bool:
        return self.pp_rank == 0

    def is_last_pp_rank(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    def prev_pp_rank(self) -> int:
        return self.pp_group[(self.pp_rank - 1) % self.pp_size]

    def next_pp_rank(self) -> int:
        return self.pp_group[(self.pp_rank + 1) % self.pp_size]

    def pp_layers(self, num_layers: int) -> List[int]:
        """Contiguous layer shard for this pp rank."""
        per = (num_layers + self.pp_size - 1) // self.pp_size
        start = self.pp_rank * per
        return list(range(start, min(start + per, num_layers)))

    def ep_experts(self, num_experts: int) -> List[int]:
        per = (num_experts + self.moe_ep_size - 1) // self.moe_ep_size
        start = self.moe_ep_rank * per
        return list(range(start, min(start + per, num_experts)))

    def __repr__(self):
        return (
            f"Mapping(world={self.world_size}, rank={self.rank}, "
            f"tp={self.tp_size}, pp={self.pp_size}, cp={self.cp_size}, "
            f"moe_tp={self.moe_tp_size}, moe_ep={self.moe_ep_size})"
        )
