"""Kimi Delta Attention surface (reference parity: flashinfer/kda.py,
kda_decode.py, kda_prefill.py — recurrent_kda, RecurrentKDAPrefillWrapper,
fused_kda_decode, packed_kda_decode). The delta-rule math lives in
csrc/gdn.hip (per-channel-gate template); this module is the KDA-named
API over it."""
from __future__ import annotations

from typing import Optional

import torch

from .gdn import chunk_kda, fused_kda_decode


def recurrent_kda(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    g: torch.Tensor, beta: torch.Tensor,
    initial_state: Optional[torch.Tensor] = None,
    output_final_state: bool = False,
    cu_seqlens: Optional[torch.Tensor] = None,
    scale: Optional[float] = None,
    **kwargs,
):
    r"""Recurrent (exact) KDA over ragged sequences — the per-channel-gate
    delta rule scanned token by token. [T, H, D] inputs with cu_seqlens,
    or [B, L, H, D] batch form."""
    batched = q.dim() == 4
    if batched:
        B, L, H, D = q.shape
        cu_seqlens = torch.arange(0, (B + 1) * L, L, dtype=torch.int32)
        q, k, v = (t.reshape(B * L, H, D) for t in (q, k, v))
        g = g.reshape(B * L, H, D)
        beta = beta.reshape(B * L, H)
    if cu_seqlens is None:
        cu_seqlens = torch.tensor([0, q.shape[0]], dtype=torch.int32)
    res = chunk_kda(q, k, v, g, beta, cu_seqlens, scale=scale,
                    initial_state=initial_state,
                    output_final_state=output_final_state)
    if batched:
        if output_final_state:
            o, s = res
            return o.reshape(B, L, H, D), s
        return res.reshape(B, L, H, D)
    return res


def packed_kda_decode(
    state: torch.Tensor,   # [B, H, D, D]
    qkv: torch.Tensor,     # [B, H, 3*D] packed q|k|v
    g: torch.Tensor,       # [B, H, D]
    beta: torch.Tensor,    # [B, H]
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""Decode step consuming the packed q|k|v projection output directly."""
    D = state.shape[-1]
    q, k, v = qkv.split([D, D, D], dim=-1)
    return fused_kda_decode(state, q.contiguous(), k.contiguous(),
                            v.contiguous(), g, beta, out)


class RecurrentKDAPrefillWorkspace:
    r"""Workspace holder (reference name parity — the CDNA4 scan kernel
    keeps its state in registers, so only the final-state buffer exists)."""

    def __init__(self, num_seqs: int, num_heads: int, head_dim: int,
                 device="cuda"):
        self.final_state = torch.empty(num_seqs, num_heads, head_dim,
                                       head_dim, dtype=torch.float32,
                                       device=device)


class RecurrentKDAPrefillWrapper:
    r"""plan/run wrapper over the KDA chunked-prefill scan."""

    def __init__(self, workspace: Optional[RecurrentKDAPrefillWorkspace] = None):
        self._ws = workspace
        self._cu = None
        self._scale = None

    def plan(self, cu_seqlens: torch.Tensor, num_heads: int, head_dim: int,
             scale: Optional[float] = None, **kwargs):
        self._cu = cu_seqlens.to(torch.int32)
        self._scale = scale

    def run(self, q, k, v, g, beta, initial_state=None,
            output_final_state: bool = False, **kwargs):
        return chunk_kda(q, k, v, g, beta, self._cu, scale=self._scale,
                         initial_state=initial_state,
                         output_final_state=output_final_state)

    forward = run
