"""mHC (multi-head hyper-connections) fused ops (reference parity:
flashinfer/mhc.py mhc_post:76, mhc_pre_big_fuse:176,
mhc_pre_big_fuse_with_prenorm:334; math per the reference's
tests/mhc/test_mhc_pre_big_fuse.py)."""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ._lib import get_ext


def mhc_post(
    x: torch.Tensor,          # [..., H]
    residual: torch.Tensor,   # [..., 4, H]
    post_layer_mix: torch.Tensor,   # [..., 4] or [..., 4, 1]
    comb_res_mix: torch.Tensor,     # [..., 4, 4]
) -> torch.Tensor:
    r"""``out[..., new, h] = x[..., h] * post_layer_mix[..., new]
    + sum_old residual[..., old, h] * comb_res_mix[..., old, new]``"""
    hc, H = residual.shape[-2], residual.shape[-1]
    if hc != 4:
        raise ValueError("mHC is hard-wired to HC=4")
    xf = x.reshape(-1, H).contiguous()
    rf = residual.reshape(-1, hc, H).contiguous()
    out = torch.empty_like(rf)
    get_ext().mhc_post(
        xf, rf, post_layer_mix.reshape(-1, hc).float().contiguous(),
        comb_res_mix.reshape(-1, hc, hc).float().contiguous(), out,
    )
    return out.reshape_as(residual)


def _mhc_pre(
    dot_mix, sqrsum, residual, mhc_scale, mhc_base, k, rms_eps, mhc_pre_eps,
    mhc_sinkhorn_eps, mhc_post_mult_value, sinkhorn_repeat, num_splits,
):
    hc, H = residual.shape[-2], residual.shape[-1]
    if hc != 4:
        raise ValueError("mHC is hard-wired to HC=4")
    outer = residual.shape[:-2]
    rf = residual.reshape(-1, hc, H).contiguous()
    tokens = rf.shape[0]
    dm = dot_mix.reshape(num_splits, tokens, 24).float().contiguous() \
        if num_splits > 1 else dot_mix.reshape(1, tokens, 24).float().contiguous()
    sq = None
    if sqrsum is not None:
        sq = (sqrsum.reshape(num_splits, tokens) if num_splits > 1
              else sqrsum.reshape(1, tokens)).float().contiguous()
        # host-side split reduction keeps the kernel's f32 order deterministic
        sq = sq.sum(0, keepdim=True) if num_splits > 1 else sq
        dm = dm.sum(0, keepdim=True) if num_splits > 1 else dm
        ns = 1
    else:
        dm = dm.sum(0, keepdim=True) if num_splits > 1 else dm
        ns = 1
    post_mix = torch.empty(tokens, hc, 1, dtype=torch.float32,
                           device=residual.device)
    comb_mix = torch.empty(tokens, hc, hc, dtype=torch.float32,
                           device=residual.device)
    layer_input = torch.empty(tokens, H, dtype=residual.dtype,
                              device=residual.device)
    get_ext().mhc_pre(
        dm.reshape(ns * tokens, 24), sq.reshape(ns * tokens) if sq is not None
        else None, rf, mhc_scale.float().contiguous(),
        mhc_base.float().contiguous(), post_mix, comb_mix, layer_input, ns,
        1.0 / float(k), rms_eps, mhc_pre_eps, mhc_sinkhorn_eps,
        mhc_post_mult_value, sinkhorn_repeat,
    )
    return (post_mix.reshape(*outer, hc, 1), comb_mix.reshape(*outer, hc, hc),
            layer_input.reshape(*outer, H))


def mhc_pre_big_fuse(
    dot_mix: torch.Tensor,   # [(splits,) ..., 24]
    sqrsum: torch.Tensor,    # [(splits,) ...]
    residual: torch.Tensor,  # [..., 4, H]
    mhc_scale: torch.Tensor,  # [3]
    mhc_base: torch.Tensor,   # [24]
    k: int,
    rms_eps: float = 1e-6,
    mhc_pre_eps: float = 1e-6,
    mhc_sinkhorn_eps: float = 1e-6,
    mhc_post_mult_value: float = 1.0,
    sinkhorn_repeat: int = 20,
    num_splits: int = 1,
    block_size: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    r"""Returns ``(post_mix [..., 4, 1], comb_mix [..., 4, 4],
    layer_input [..., H])`` — sigmoid pre/post mixes, Sinkhorn-normalized
    4x4 residual-combination matrix, and the pre-mixed layer input."""
    if num_splits not in (1, 2, 4, 8, 16):
        raise ValueError("num_splits must be one of {1, 2, 4, 8, 16}")
    return _mhc_pre(dot_mix, sqrsum, residual, mhc_scale, mhc_base, k, rms_eps,
                    mhc_pre_eps, mhc_sinkhorn_eps, mhc_post_mult_value,
                    sinkhorn_repeat, num_splits)


def mhc_pre_big_fuse_with_prenorm(
    dot_mix: torch.Tensor,
    residual: torch.Tensor,
    mhc_scale: torch.Tensor,
    mhc_base: torch.Tensor,
    rms_eps: float = 1e-6,
    mhc_pre_eps: float = 1e-6,
    mhc_sinkhorn_eps: float = 1e-6,
    mhc_post_mult_value: float = 1.0,
    sinkhorn_repeat: int = 20,
    block_size: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    r"""Same as :func:`mhc_pre_big_fuse` but the residual square-sum is
    computed in-kernel over the flattened [4*H] residual (k = 4*H)."""
    if dot_mix.dim() == residual.dim():
        dot_mix = dot_mix.squeeze(0)
    hc, H = residual.shape[-2], residual.shape[-1]
    return _mhc_pre(dot_mix, None, residual, mhc_scale, mhc_base, hc * H,
                    rms_eps, mhc_pre_eps, mhc_sinkhorn_eps, mhc_post_mult_value,
                    sinkhorn_repeat, 1)
