"""Fused Mixture-of-Experts forward (reference parity: flashinfer/fused_moe/
core.py cutlass_fused_moe:1180 pipeline — routing -> expand/permute by expert
-> grouped GEMM1 (gated) -> activation -> grouped GEMM2 -> finalize
scatter-reduce) on the CDNA4 grouped MFMA GEMM.

Also the routing functions (top-k softmax, DeepSeek-V3 no-aux-loss grouped
top-k with sigmoid scores — reference trtllm_fused_moe_routing_deepseek /
noAuxTcKernels).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ._lib import get_ext
from .activation import silu_and_mul, gelu_and_mul
from .utils import ceil_div


def moe_topk_softmax(
    router_logits: torch.Tensor, top_k: int, renormalize: bool = True
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Standard Mixtral-style routing: softmax then top-k (+ renorm)."""
    probs = torch.softmax(router_logits.float(), dim=-1)
    weights, ids = torch.topk(probs, top_k, dim=-1)
    if renormalize:
        weights = weights / weights.sum(dim=-1, keepdim=True).clamp(min=1e-20)
    return weights, ids.to(torch.int32)


def dsv3_routing(
    router_logits: torch.Tensor, top_k: int, n_group: int, topk_group: int,
    routed_scaling_factor: float = 1.0, bias: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """DeepSeek-V3 no-aux-loss routing: sigmoid scores (+bias for selection),
    group-limited top-k (keep topk_group groups by their top-2 sum), weights
    from the unbiased scores, normalized and scaled."""
    scores = torch.sigmoid(router_logits.float())
    sel = scores + bias.float() if bias is not None else scores
    T, E = scores.shape
    gsize = E // n_group
    gsel = sel.view(T, n_group, gsize)
    group_score = gsel.topk(2, dim=-1).values.sum(-1)  # [T, n_group]
    keep_groups = group_score.topk(topk_group, dim=-1).indices
    mask = torch.zeros(T, n_group, dtype=torch.bool, device=scores.device)
    mask.scatter_(1, keep_groups, True)
    sel_masked = sel.masked_fill(
        ~mask.unsqueeze(-1).expand(T, n_group, gsize).reshape(T, E), float("-inf")
    )
    ids = sel_masked.topk(top_k, dim=-1).indices
    w = scores.gather(1, ids)
    w = w / w.sum(-1, keepdim=True).clamp(min=1e-20) * routed_scaling_factor
    return w, ids.to(torch.int32)


def _permute(x: torch.Tensor, topk_ids: torch.Tensor, num_experts: int):
    T, k = topk_ids.shape
    flat = topk_ids.reshape(-1).long()
    order = torch.argsort(flat, stable=True)
    counts = torch.bincount(flat, minlength=num_experts)
    m_indptr = torch.zeros(num_experts + 1, dtype=torch.int32, device=x.device)
    m_indptr[1:] = counts.cumsum(0).to(torch.int32)
    token_of_copy = order // k  # source token per permuted row
    a_perm = x.index_select(0, token_of_copy)
    return a_perm, m_indptr, order, token_of_copy


def fused_moe(
    x: torch.Tensor,
    w13: torch.Tensor,  # [E, 2*inter, hidden]  (gate | up rows)
    w2: torch.Tensor,   # [E, hidden, inter]
    topk_weights: torch.Tensor,  # [T, k] f32
    topk_ids: torch.Tensor,      # [T, k] int
    activation: str = "silu",
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    T, H = x.shape
    E, I2, Hw = w13.shape
    assert Hw == H, "w13 must be [E, 2*inter, hidden]"
    inter = I2 // 2
    k = topk_ids.shape[1]
    ext = get_ext()

    a_perm, m_indptr, order, token_of_copy = _permute(x, topk_ids, E)
    R = a_perm.shape[0]
    max_m_tiles = ceil_div(R, 128) + 1

    h1 = torch.empty(R, I2, dtype=x.dtype, device=x.device)
    ext.group_gemm_nt(a_perm, w13, h1, m_indptr, None, max_m_tiles)
    act = {"silu": silu_and_mul, "gelu": gelu_and_mul}[activation](h1)
    h2 = torch.empty(R, H, dtype=x.dtype, device=x.device)
    ext.group_gemm_nt(act, w2, h2, m_indptr, None, max_m_tiles)

    # finalize: out[token] = sum_j weight[t, j] * h2[row of (t, j)]
    inv = torch.empty_like(order)
    inv[order] = torch.arange(R, device=x.device)
    contrib = h2.index_select(0, inv).view(T, k, H)
    res = (contrib.float() * topk_weights.float().unsqueeze(-1)).sum(1).to(x.dtype)
    if out is not None:
        out.copy_(res)
        return out
    return res


def cutlass_fused_moe(
    input: torch.Tensor,
    token_selected_experts: torch.Tensor,
    token_final_scales: torch.Tensor,
    fc1_expert_weights: torch.Tensor,
    fc2_expert_weights: torch.Tensor,
    output_dtype: torch.dtype = torch.bfloat16,
    quant_scales=None,
    **kwargs,
) -> torch.Tensor:
    r"""Reference-compatible alias (flashinfer fused_moe/core.py:1180 arg
    order): fc1 weights [E, 2*inter, hidden], fc2 [E, hidden, inter]."""
    return fused_moe(
        input, fc1_expert_weights, fc2_expert_weights,
        token_final_scales, token_selected_experts,
    ).to(output_dtype)
