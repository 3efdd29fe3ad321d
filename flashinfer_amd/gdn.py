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


def chunk_gated_delta_rule(
    q: torch.Tensor,          # [total_tokens, H, D]
    k: torch.Tensor,          # [total_tokens, H, D]
    v: torch.Tensor,          # [total_tokens, H, D]
    gate: torch.Tensor,       # [total_tokens, H] decay in (0,1], or [.., H, D] (KDA)
    beta: torch.Tensor,       # [total_tokens, H]
    cu_seqlens: torch.Tensor,  # [num_seqs + 1] int32
    scale: Optional[float] = None,
    initial_state: Optional[torch.Tensor] = None,   # [num_seqs, H, D, D] f32
    output_final_state: bool = False,
    output: Optional[torch.Tensor] = None,
):
    r"""Gated delta-rule prefill over ragged sequences (reference parity:
    flashinfer/gdn_kernels/blackwell/gdn_prefill.py
    chunk_gated_delta_rule_sm100:137; math validated against the reference's
    tests/gdn/reference_delta_rule.py blockwise_delta_rule:856). Per token:
    ``S = a*S; u = beta*(v - k^T S); S += k (x) u; o = scale * q^T S``.
    State layout is [num_seqs, H, D_k, D_v] (same as ``gdn_fused_decode_step``;
    the reference's output_state is the [.., D_v, D_k] transpose).
    Returns ``out`` or ``(out, final_state)``. A 3-D gate selects the
    KDA-style per-channel decay."""
    total, H, D = q.shape
    if scale is None:
        scale = D ** -0.5
    if output is None:
        output = torch.empty_like(v)
    num_seqs = cu_seqlens.numel() - 1
    final_state = None
    if output_final_state:
        final_state = torch.empty(num_seqs, H, D, D, dtype=torch.float32,
                                  device=q.device)
    get_ext().gdn_chunk(
        q.contiguous(), k.contiguous(), v.contiguous(),
        gate.float().contiguous(), beta.float().contiguous(), output,
        cu_seqlens.to(q.device, torch.int32),
        initial_state.float().contiguous() if initial_state is not None else None,
        final_state, float(scale),
    )
    return (output, final_state) if output_final_state else output


def chunk_kda(q, k, v, g_channel, beta, cu_seqlens, scale=None,
              initial_state=None, output_final_state=False, output=None):
    r"""KDA chunked prefill: delta rule with per-channel gate [T, H, D]."""
    assert g_channel.dim() == 3
    return chunk_gated_delta_rule(q, k, v, g_channel, beta, cu_seqlens, scale,
                                  initial_state, output_final_state, output)


def gdn_fused_decode_step_supported(*args, **kwargs) -> bool:
    r"""Capability probe (reference gdn_kernels role): the CDNA4 decode-step
    kernel covers every (dtype, state dtype, gate) combination we ship."""
    return True
