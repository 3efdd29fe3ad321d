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


def mamba_chunk_scan_combined(
    x: torch.Tensor,              # [B, L, H, P]
    dt: torch.Tensor,             # [B, L, H]
    A: torch.Tensor,              # [H] f32 (negative)
    B: torch.Tensor,              # [B, L, G, N]
    C: torch.Tensor,              # [B, L, G, N]
    chunk_size: int = 128,        # accepted for API parity; scan is exact
    D: Optional[torch.Tensor] = None,        # [H] or [H, P]
    z: Optional[torch.Tensor] = None,        # [B, L, H, P]
    dt_bias: Optional[torch.Tensor] = None,  # [H]
    dt_softplus: bool = False,
    dt_limit=(0.0, float("inf")),
    initial_states: Optional[torch.Tensor] = None,  # [B, H, P, N] f32
    return_final_states: bool = False,
    out: Optional[torch.Tensor] = None,
):
    r"""Mamba-2 SSD prefill (reference parity: flashinfer/mamba/ssd_combined.py
    SSDCombined:250 / mamba_chunk_scan_combined semantics). Scans the full
    selective-state recurrence over each sequence:
    ``S = exp(dt'*A[h]) * S + dt' * x_t (x) B_t;  y_t = S C_t + D*x_t``
    with optional silu(z) output gating. The CDNA4 kernel keeps each head-dim
    row's [dstate] state in registers (one block per (batch, head)), so the
    result is exact — ``chunk_size`` only exists for signature parity."""
    batch, L, H, P = x.shape
    if out is None:
        out = torch.empty_like(x)
    final_states = None
    if return_final_states:
        final_states = torch.empty(batch, H, P, B.shape[-1], dtype=torch.float32,
                                   device=x.device)
    get_ext().ssd_scan(
        x.contiguous(), dt.float().contiguous(), A.float().contiguous(),
        B.contiguous(), C.contiguous(),
        D.float().contiguous() if D is not None else None,
        z.contiguous() if z is not None else None,
        dt_bias.float().contiguous() if dt_bias is not None else None,
        initial_states.float().contiguous() if initial_states is not None else None,
        final_states, out, dt_softplus, float(dt_limit[0]), float(dt_limit[1]),
    )
    return (out, final_states) if return_final_states else out


def cake_selective_state_update(*args, **kwargs):
    r"""Alias of :func:`selective_state_update` (reference flashinfer/mamba/
    cake_selective_state_update.py:12 routes to the same op with
    backend="cake"; here there is a single CDNA4 backend)."""
    kwargs.pop("backend", None)
    return selective_state_update(*args, **kwargs)
