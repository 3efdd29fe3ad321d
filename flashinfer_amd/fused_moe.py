"""Fused Mixture-of-Experts forward (reference parity: flashinfer/fused_moe/
core.py cutlass_fused_moe:1180 pipeline — routing -> expand/permute by expert
-> grouped GEMM1 (gated) -> activation -> grouped GEMM2 -> finalize
scatter-reduce) on the CDNA4 grouped MFMA GEMM.

Also the routing functions (top-k softmax, DeepSeek-V3 no-aux-loss grouped
top-k with sigmoid scores — reference trtllm_fused_moe_routing_deepseek /
noAuxTcKernels).
"""
from __future__ import annotations

import enum

from typing import Optional, Tuple

import torch

from ._lib import get_ext
from .activation import silu_and_mul, gelu_and_mul
from .utils import ceil_div


def moe_topk_softmax(
    router_logits: torch.Tensor, top_k: int, renormalize: bool = True
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Standard Mixtral-style routing: softmax then top-k (+ renorm).

    One fused wave-per-token kernel on GPU (reference
    trtllm_gen_routing topk-softmax role); torch fallback on CPU."""
    T, E = router_logits.shape
    if router_logits.is_cuda and E <= 256 and top_k <= 16:
        logits = router_logits.float().contiguous()
        weights = torch.empty(T, top_k, dtype=torch.float32,
                              device=logits.device)
        ids = torch.empty(T, top_k, dtype=torch.int32, device=logits.device)
        get_ext().moe_topk_softmax_run(logits, weights, ids, renormalize)
        return weights, ids
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
    from the unbiased scores, normalized and scaled.

    Fused wave-per-token kernel on GPU (reference
    trtllm_fused_moe_routing_deepseek.cu / noAuxTcKernels role)."""
    T, E = router_logits.shape
    if (router_logits.is_cuda and E <= 256 and top_k <= 16
            and n_group <= 32 and E // n_group <= 32):
        logits = router_logits.float().contiguous()
        weights = torch.empty(T, top_k, dtype=torch.float32,
                              device=logits.device)
        ids = torch.empty(T, top_k, dtype=torch.int32, device=logits.device)
        get_ext().dsv3_routing_run(
            logits, bias.float().contiguous() if bias is not None else None,
            weights, ids, n_group, topk_group, routed_scaling_factor)
        return weights, ids
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


def _grouped_nt_bf16(a, w, m_indptr, max_m_tiles):
    """bf16 grouped NT GEMM: hipBLASLt's grouped path (torch._grouped_mm,
    1219 TF measured vs 797 for the in-house kernel — profiles/
    r01_gemm_ab.txt) with the in-house kernel as fallback."""
    if hasattr(torch, "_grouped_mm"):
        try:
            return torch._grouped_mm(a, w.transpose(1, 2),
                                     offs=m_indptr[1:].to(torch.int32))
        except Exception:
            pass
    out = torch.zeros(a.shape[0], w.shape[1], dtype=torch.bfloat16,
                      device=a.device)
    get_ext().group_gemm_nt(a, w, out, m_indptr, None, max_m_tiles)
    return out


def _build_permute(topk_ids: torch.Tensor, num_experts: int, align: int = 1):
    """hist -> scan -> atomic scatter kernels (replaces the torch
    argsort/bincount/cumsum chain — reference moe_kernels.h expand stage).
    Returns (m_indptr, token_of_copy, inv) where inv[t, j] is the permuted
    row of copy (t, j). Atomic order within an expert is arbitrary, but all
    downstream ops are row-local and the finalize gathers by inv, so outputs
    are bitwise deterministic."""
    T, k = topk_ids.shape
    dev = topk_ids.device
    ids = topk_ids.to(torch.int32).contiguous()
    rmax = T * k + (num_experts * (align - 1) if align > 1 else 0)
    counts = torch.zeros(num_experts, dtype=torch.int32, device=dev)
    cursor = torch.empty(num_experts, dtype=torch.int32, device=dev)
    m_indptr = torch.empty(num_experts + 1, dtype=torch.int32, device=dev)
    # pad rows gather token 0 (valid data, outputs never read) — zeros, not
    # empty, so no NaN can enter the pipeline
    token_of_copy = torch.zeros(rmax, dtype=torch.int32, device=dev)
    inv = torch.empty(T * k, dtype=torch.int32, device=dev)
    get_ext().moe_build_permute_run(ids, counts, m_indptr, cursor,
                                    token_of_copy, inv, align)
    return m_indptr, token_of_copy, inv


def _permute(x: torch.Tensor, topk_ids: torch.Tensor, num_experts: int):
    m_indptr, token_of_copy, inv = _build_permute(topk_ids, num_experts)
    a_perm = torch.empty(topk_ids.numel(), x.shape[1], dtype=x.dtype,
                         device=x.device)
    get_ext().gather_rows(x, a_perm, token_of_copy[: topk_ids.numel()])
    return a_perm, m_indptr, inv, token_of_copy


def fused_moe(
    x: torch.Tensor,
    w13: torch.Tensor,  # [E, 2*inter, hidden]  (gate | up rows; bf16 or fp8)
    w2: torch.Tensor,   # [E, hidden, inter]
    topk_weights: torch.Tensor,  # [T, k] f32
    topk_ids: torch.Tensor,      # [T, k] int
    activation: str = "silu",
    out: Optional[torch.Tensor] = None,
    w13_scale: Optional[torch.Tensor] = None,  # [E, H/128, 2I/128] (fp8 path)
    w2_scale: Optional[torch.Tensor] = None,   # [E, I/128, H/128]
) -> torch.Tensor:
    r"""bf16 path: grouped bf16 MFMA GEMMs. fp8 path (weights in e4m3 +
    128x128 block scales): activations are per-128-group quantized on the
    fly and the grouped fp8 MFMA GEMM applies groupwise rescaling (the
    reference's trtllm_fp8_block_scale_moe contract)."""
    T, H = x.shape
    E, I2, Hw = w13.shape
    assert Hw == H, "w13 must be [E, 2*inter, hidden]"
    inter = I2 // 2
    k = topk_ids.shape[1]
    ext = get_ext()
    fp8 = w13.dtype == torch.float8_e4m3fn

    R = T * k
    max_m_tiles = ceil_div(R, 128) + 1

    if fp8:
        # fused data movers: gather+quant feeds GEMM1, silu_mul+quant feeds
        # GEMM2 — the bf16 intermediate round-trips are gone. At high expert
        # counts the segments are 128-aligned and the grouped GEMM runs a
        # FLAT tile grid (the z-grid pads every expert to the worst-case
        # tile count: E x total_tiles dead dispatches); at small E the pad
        # rows cost more than the dead WGs (measured: E=8 1.455 -> 1.405
        # M tok/s), so the z-grid stays.
        use_flat = E >= 16
        # MX mode (u8 e8m0 weight scales from per_block_quant_mxfp8): the
        # hardware-scaled f8f6f4 MFMA applies the scales — no rescale VALU
        mx = w13_scale.dtype == torch.uint8
        m_indptr, token_of_copy, inv = _build_permute(
            topk_ids, E, align=128 if use_flat else 1)
        Rp = token_of_copy.shape[0]
        flat_tiles = Rp // 128 if use_flat else 0
        a_q = torch.empty(Rp, H, dtype=torch.uint8, device=x.device)
        a_s = (torch.empty(Rp, H // 128, dtype=torch.uint8, device=x.device)
               if mx else
               torch.empty(H // 128, Rp, dtype=torch.float32, device=x.device))
        ext.gather_quant_run(x, token_of_copy, a_q, a_s)
        h1 = torch.empty(Rp, I2, dtype=torch.bfloat16, device=x.device)
        ext.gemm_fp8_grouped(a_q, w13.view(torch.uint8), h1,
                             m_indptr, None, max_m_tiles, a_s,
                             w13_scale.contiguous(), 1.0, flat_tiles)
        act_q = torch.empty(Rp, inter, dtype=torch.uint8, device=x.device)
        act_s = (torch.empty(Rp, inter // 128, dtype=torch.uint8,
                             device=x.device)
                 if mx else
                 torch.empty(inter // 128, Rp, dtype=torch.float32,
                             device=x.device))
        ext.silu_mul_quant_run(h1, act_q, act_s, activation == "gelu")
        h2 = torch.empty(Rp, H, dtype=torch.bfloat16, device=x.device)
        ext.gemm_fp8_grouped(act_q, w2.view(torch.uint8), h2,
                             m_indptr, None, max_m_tiles, act_s,
                             w2_scale.contiguous(), 1.0, flat_tiles)
    else:
        a_perm, m_indptr, inv, token_of_copy = _permute(x, topk_ids, E)
        h1 = _grouped_nt_bf16(a_perm, w13, m_indptr, max_m_tiles)
        act = {"silu": silu_and_mul, "gelu": gelu_and_mul}[activation](h1)
        h2 = _grouped_nt_bf16(act, w2, m_indptr, max_m_tiles)

    # finalize: out[token] = sum_j weight[t, j] * h2[row of (t, j)]
    res = out if out is not None else torch.empty(T, H, dtype=x.dtype, device=x.device)
    get_ext().moe_finalize(h2, res, inv.view(T, k),
                           topk_weights.float().contiguous())
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


def trtllm_fp8_block_scale_moe(
    routing_logits: torch.Tensor,
    routing_bias: Optional[torch.Tensor],
    hidden_states: torch.Tensor,
    gemm1_weights: torch.Tensor,       # [E, 2*inter, hidden] fp8
    gemm1_weights_scale: torch.Tensor, # [E, hidden/128, 2*inter/128]
    gemm2_weights: torch.Tensor,       # [E, hidden, inter] fp8
    gemm2_weights_scale: torch.Tensor, # [E, inter/128, hidden/128]
    num_experts: int, top_k: int,
    n_group: Optional[int] = None, topk_group: Optional[int] = None,
    intermediate_size: Optional[int] = None,
    routed_scaling_factor: float = 1.0,
    routing_method_type: int = 0,
    **kwargs,
) -> torch.Tensor:
    r"""Reference-compatible fp8 block-scale MoE (flashinfer
    trtllm_fp8_block_scale_moe): DSv3 grouped routing when n_group is set,
    renormalized top-k softmax otherwise."""
    if n_group:
        w, ids = dsv3_routing(routing_logits, top_k, n_group, topk_group,
                              routed_scaling_factor, bias=routing_bias)
    else:
        w, ids = moe_topk_softmax(routing_logits, top_k)
    return fused_moe(hidden_states, gemm1_weights, gemm2_weights, w, ids,
                     w13_scale=gemm1_weights_scale, w2_scale=gemm2_weights_scale)


class RoutingMethodType(enum.IntEnum):
    """Reference trtllm routing-method enum (fused_moe/core.py role)."""
    Default = 0            # softmax -> top-k
    Renormalize = 1        # top-k -> softmax over the k
    DeepSeekV3 = 2         # grouped sigmoid (dsv3_routing)
    Llama4 = 3             # top-1 sigmoid
    RenormalizeNaive = 4
    TopK = 5


class ActivationType(enum.IntEnum):
    Swiglu = 0
    Geglu = 1


def is_gated_activation(act) -> bool:
    return act in (ActivationType.Swiglu, ActivationType.Geglu, "silu", "gelu")


def _route(router_logits, top_k, method: int, routed_scaling_factor=1.0,
           n_group=1, topk_group=1, routing_bias=None):
    logits = router_logits.float()
    if method == RoutingMethodType.DeepSeekV3:
        return dsv3_routing(logits, top_k, n_group, topk_group,
                            routed_scaling_factor, bias=routing_bias)
    if method == RoutingMethodType.Llama4:
        w, ids = torch.topk(torch.sigmoid(logits), 1, dim=-1)
        return w * routed_scaling_factor, ids.int()
    if method in (RoutingMethodType.Renormalize,
                  RoutingMethodType.RenormalizeNaive):
        vals, ids = torch.topk(logits, top_k, dim=-1)
        return torch.softmax(vals, -1) * routed_scaling_factor, ids.int()
    w, ids = torch.topk(torch.softmax(logits, -1), top_k, dim=-1)
    return w * routed_scaling_factor, ids.int()


def trtllm_bf16_moe(
    routing_logits: torch.Tensor, hidden_states: torch.Tensor,
    gemm1_weights: torch.Tensor, gemm2_weights: torch.Tensor,
    top_k: int, routing_method_type: int = RoutingMethodType.Renormalize,
    routed_scaling_factor: float = 1.0, n_group: int = 1, topk_group: int = 1,
    routing_bias=None, **kwargs,
) -> torch.Tensor:
    r"""Routed bf16 MoE (reference trtllm_bf16_moe contract): routing +
    grouped GEMM pipeline in one call."""
    w, ids = _route(routing_logits, top_k, routing_method_type,
                    routed_scaling_factor, n_group, topk_group, routing_bias)
    return fused_moe(hidden_states, gemm1_weights, gemm2_weights, w, ids)


def trtllm_bf16_routed_moe(hidden_states, topk_ids, topk_weights,
                           gemm1_weights, gemm2_weights, **kwargs):
    r"""Pre-routed variant: caller supplies (ids, weights)."""
    return fused_moe(hidden_states, gemm1_weights, gemm2_weights,
                     topk_weights, topk_ids)


def trtllm_fp8_per_tensor_scale_moe(
    routing_logits: torch.Tensor, routing_bias, hidden_states: torch.Tensor,
    gemm1_weights: torch.Tensor, output1_scales_scalar,
    gemm2_weights: torch.Tensor, output2_scales_scalar, num_experts: int,
    top_k: int, n_group: int = 1, topk_group: int = 1,
    intermediate_size: int = 0, local_expert_offset: int = 0,
    local_num_experts: int = 0, routed_scaling_factor: float = 1.0,
    routing_method_type: int = RoutingMethodType.Renormalize, **kwargs,
) -> torch.Tensor:
    r"""Routed fp8 MoE with per-tensor weight scales folded into the
    groupwise pipeline's block scales."""
    w, ids = _route(routing_logits, top_k, routing_method_type,
                    routed_scaling_factor, n_group, topk_group, routing_bias)
    E = gemm1_weights.shape[0]
    N1, H = gemm1_weights.shape[1], gemm1_weights.shape[2]
    N2 = gemm2_weights.shape[1]
    dev = hidden_states.device

    def scales(n, k, s):
        t = torch.as_tensor(s, dtype=torch.float32, device=dev).reshape(-1)
        base = torch.ones(E, (k + 127) // 128, (n + 127) // 128,
                          dtype=torch.float32, device=dev)
        return base * t.view(-1, 1, 1)

    return fused_moe(
        hidden_states, gemm1_weights, gemm2_weights, w, ids,
        w13_scale=scales(N1, H, output1_scales_scalar),
        w2_scale=scales(N2, N1 // 2, output2_scales_scalar))


def trtllm_fp8_per_tensor_scale_routed_moe(hidden_states, topk_ids,
                                           topk_weights, gemm1_weights,
                                           s1, gemm2_weights, s2, **kwargs):
    E = gemm1_weights.shape[0]
    N1, H = gemm1_weights.shape[1], gemm1_weights.shape[2]
    N2 = gemm2_weights.shape[1]
    dev = hidden_states.device

    def scales(n, k, s):
        t = torch.as_tensor(s, dtype=torch.float32, device=dev).reshape(-1)
        base = torch.ones(E, (k + 127) // 128, (n + 127) // 128,
                          dtype=torch.float32, device=dev)
        return base * t.view(-1, 1, 1)

    return fused_moe(hidden_states, gemm1_weights, gemm2_weights,
                     topk_weights, topk_ids,
                     w13_scale=scales(N1, H, s1), w2_scale=scales(N2, N1 // 2, s2))


def trtllm_fp8_block_scale_routed_moe(hidden_states, topk_ids, topk_weights,
                                      gemm1_weights, w13_scale, gemm2_weights,
                                      w2_scale, **kwargs):
    r"""Pre-routed variant of trtllm_fp8_block_scale_moe."""
    return fused_moe(hidden_states, gemm1_weights, gemm2_weights,
                     topk_weights, topk_ids, w13_scale=w13_scale,
                     w2_scale=w2_scale)
