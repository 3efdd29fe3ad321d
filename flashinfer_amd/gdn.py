"""Gated DeltaNet ops (reference parity: flashinfer/gdn_kernels
gdn_fused_decode_step — the serving decode step of the gated delta rule used
by Qwen3-Next-style hybrid models; chunked prefill arrives in a later drop)."""
from __future__ import annotations

from typing import Optional

import torch

from ._lib import get_ext


def gdn_fused_decode_step(
    state: torch.Tensor,  # [B, H, Dk, Dv] in/out (f32 or activation dtype)
    q: torch.Tensor,      # [B, H, Dk]
    k: torch.Tensor,      # [B, H, Dk]
    v: torch.Tensor,      # [B, H, Dv]
    g: torch.Tensor,      # [B, H] f32 gate decay, or [B, H, Dk] per-channel (KDA)
    beta: torch.Tensor,   # [B, H] f32 delta-rule step size
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""One gated-delta-rule step (updates state in place, returns o = q^T S):
    ``S = g*S + k (x) (beta * (v - k^T (g*S)));  o = q^T S``.
    A 3-D gate enables the KDA-style per-channel (diagonal) decay."""
    if out is None:
        out = torch.empty_like(v)
    get_ext().gdn_decode(state, q.contiguous(), k.contiguous(), v.contiguous(),
                         g.float().contiguous(), beta.float().contiguous(), out)
    return out


def fused_kda_decode(state, q, k, v, g_channel, beta, out=None):
    r"""Kimi Delta Attention decode step (reference flashinfer/kda
    fused_kda_decode role): delta rule with per-channel gate [B, H, Dk]."""
    assert g_channel.dim() == 3
    return gdn_fused_decode_step(state, q, k, v, g_channel, beta, out)
