"""Mamba/SSM ops (reference parity: flashinfer/mamba/ —
selective_state_update decode step, checkpointing SSU snapshot/rollback for
speculative decoding)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext


def selective_state_update(
    state: torch.Tensor,      # [B, H, P, S] in/out
    x: torch.Tensor,          # [B, H, P]
    dt: torch.Tensor,         # [B, H]
    A: torch.Tensor,          # [H]
    B: torch.Tensor,          # [B, G, S]
    C: torch.Tensor,          # [B, G, S]
    D: Optional[torch.Tensor] = None,       # [H]
    z: Optional[torch.Tensor] = None,       # [B, H, P]
    dt_bias: Optional[torch.Tensor] = None, # [H]
    dt_softplus: bool = False,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""One SSM decode step; updates ``state`` in place and returns y [B,H,P]."""
    if out is None:
        out = torch.empty_like(x)
    get_ext().selective_state_update(
        state, x.contiguous(), dt.contiguous(), A.contiguous(), B.contiguous(),
        C.contiguous(), D, z, dt_bias, out, dt_softplus,
    )
    return out


def ssu_checkpoint(state: torch.Tensor, snapshot: torch.Tensor) -> None:
    r"""Snapshot SSM states before speculative steps (reference
    checkpointing_ssu role)."""
    snapshot.copy_(state)


def ssu_rollback(state: torch.Tensor, snapshot: torch.Tensor,
                 mask: Optional[torch.Tensor] = None) -> None:
    r"""Roll back rejected speculative steps; mask [B] selects rows to restore."""
    if mask is None:
        state.copy_(snapshot)
    else:
        state[mask] = snapshot[mask]
