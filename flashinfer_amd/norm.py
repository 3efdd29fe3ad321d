"""RMSNorm / LayerNorm ops (reference parity: flashinfer/norm/__init__.py)."""
from __future__ import annotations

import functools

from typing import Optional

import torch

from ._lib import get_ext
from .api_logging import flashinfer_api
from .fi_trace import fi_trace


def _flat2d(x: torch.Tensor) -> torch.Tensor:
    return x.reshape(-1, x.shape[-1])


@flashinfer_api
@fi_trace
def rmsnorm(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,  # accepted for API parity; no-op on MI355X
) -> torch.Tensor:
    r"""Root-mean-square normalization: ``out = input / rms(input) * weight``."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().rmsnorm(_flat2d(input), weight, _flat2d(out), eps, False)
    return out


def gemma_rmsnorm(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    enable_pdl: Optional[bool] = None,
) -> torch.Tensor:
    r"""Gemma-style RMSNorm: ``out = input / rms(input) * (weight + 1)``."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().rmsnorm(_flat2d(input), weight, _flat2d(out), eps, True)
    return out


@flashinfer_api
@fi_trace
def fused_add_rmsnorm(
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""In-place: ``residual += input; input = rmsnorm(residual) * weight``."""
    get_ext().fused_add_rmsnorm(_flat2d(input), _flat2d(residual), weight, eps, False)


def gemma_fused_add_rmsnorm(
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    get_ext().fused_add_rmsnorm(_flat2d(input), _flat2d(residual), weight, eps, True)


def layernorm(
    input: torch.Tensor,
    gemma: torch.Tensor,
    beta: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""LayerNorm over the last dimension."""
    if out is None:
        out = torch.empty_like(input)
    get_ext().layernorm(_flat2d(input), gemma, beta, _flat2d(out), eps)
    return out


def _scale_tensor(scale, device) -> torch.Tensor:
    if isinstance(scale, torch.Tensor):
        return scale.to(device, torch.float32)
    return torch.tensor([float(scale)], dtype=torch.float32, device=device)


@flashinfer_api
@fi_trace
def rmsnorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    weight: torch.Tensor,
    scale,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused RMSNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py rmsnorm_quant:214):
    ``out = ((input / rms(input)) * weight / scale).to(fp8_e4m3)``.
    ``scale`` is a float or a shape-(1,) f32 tensor read on device."""
    get_ext().rmsnorm_quant(_flat2d(input), None, weight, _flat2d(out),
                            _scale_tensor(scale, input.device), eps)


@flashinfer_api
@fi_trace
def fused_add_rmsnorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    residual: torch.Tensor,
    weight: torch.Tensor,
    scale,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused residual-add + RMSNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py fused_add_rmsnorm_quant:323). In place:
    ``residual += input; out = (rmsnorm(residual) * weight / scale).to(fp8)``."""
    get_ext().rmsnorm_quant(_flat2d(input), _flat2d(residual), weight,
                            _flat2d(out), _scale_tensor(scale, input.device), eps)


@flashinfer_api
@fi_trace
def fused_rmsnorm_silu(
    input: torch.Tensor,
    weight: torch.Tensor,
    eps: float = 1e-6,
    out: Optional[torch.Tensor] = None,
    block_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    r"""``out = silu(rmsnorm(input) * weight)`` (reference parity:
    flashinfer/norm/__init__.py fused_rmsnorm_silu:689; the fp4 block_scale
    output is N/A on CDNA4)."""
    if block_scale is not None:
        raise NotImplementedError("fp4 block scales are N/A on CDNA4")
    if out is None:
        out = torch.empty_like(input)
    assert out.dtype == input.dtype, "out dtype must match input on CDNA4 path"
    get_ext().rmsnorm_silu(_flat2d(input), weight, _flat2d(out), eps)
    return out


@flashinfer_api
@fi_trace
def layernorm_quant(
    out: torch.Tensor,
    input: torch.Tensor,
    weight: torch.Tensor,
    scale,
    bias: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
    enable_pdl: Optional[bool] = None,
) -> None:
    r"""Fused LayerNorm + fp8 quantization (reference parity:
    flashinfer/norm/__init__.py layernorm_quant role):
    ``out = (((input - mean) / sqrt(var + eps)) * weight (+ bias)
    / scale).to(fp8_e4m3)``."""
    get_ext().layernorm_quant(_flat2d(input), weight, bias, _flat2d(out),
                              _scale_tensor(scale, input.device), eps)


@functools.lru_cache(maxsize=8)
def _rope3d_cache(ppf: int, pph: int, ppw: int, head_dim: int, base: float,
                  device_str: str):
    """Interleaved 3D (frame/height/width) rope tables, WAN-style split:
    h_dim = w_dim = 2*(head_dim//6), t_dim = remainder."""
    device = torch.device(device_str)
    h_dim = w_dim = 2 * (head_dim // 6)
    t_dim = head_dim - h_dim - w_dim

    def table(dim, length):
        inv = 1.0 / (base ** (torch.arange(0, dim, 2, device=device,
                                           dtype=torch.float64) / dim))
        pos = torch.arange(length, device=device, dtype=torch.float64)
        fr = torch.einsum("i,j->ij", pos, inv)
        cos = torch.repeat_interleave(torch.cos(fr), 2, dim=-1)
        sin = torch.repeat_interleave(torch.sin(fr), 2, dim=-1)
        return cos, sin

    mx = max(ppf, pph, ppw)
    tc, ts = table(t_dim, mx)
    hc, hs = table(h_dim, mx)
    wc, ws = table(w_dim, mx)
    S = ppf * pph * ppw
    tok = torch.arange(S, device=device)
    pt, ph, pw = tok // (pph * ppw), (tok // ppw) % pph, tok % ppw
    cos = torch.cat([tc[pt], hc[ph], wc[pw]], dim=-1).float()
    sin = torch.cat([ts[pt], hs[ph], ws[pw]], dim=-1).float()
    return cos, sin  # [S, head_dim]


def fused_qk_rmsnorm_rope(
    qkv: torch.Tensor,           # [B, S, (Hq+Hk+Hv)*D]
    q_weight: torch.Tensor,      # [D] (or [Hq*D] per-hidden)
    k_weight: torch.Tensor,
    *,
    ppf: int, pph: int, ppw: int,
    num_frame_channels: int = 0, num_height_channels: int = 0,
    num_width_channels: int = 0,
    num_heads_q: int = 0, num_heads_k: int = 0, num_heads_v: int = 0,
    head_dim: int = 128,
    eps: float = 1e-6,
    base: float = 10000.0,
    interleave: bool = True,
    factor: float = 1.0, low: float = 0.0, high: float = 0.0,
    attention_factor: float = 1.0,
    is_qk_norm: bool = True,
    output_fp8: bool = False, output_quant_scale: float = 1.0,
    v_quant_scale: float = 1.0,
    q_out=None, k_out=None, v_out=None,
):
    r"""Video-DiT QKV epilogue (reference parity: flashinfer/norm/__init__.py
    fused_qk_rmsnorm_rope:1608; math per the reference's own
    tests/norm/test_fused_qk_rmsnorm_rope.py reference_qk_norm_rope:213):
    per-head RMSNorm on q/k then interleaved 3D (frame, height, width) RoPE
    over the ppf x pph x ppw token grid. NeoX element mapping is kernel-
    specific on the reference side and not supported here."""
    if not interleave:
        raise NotImplementedError("neox element mapping not supported")
    B, S, _ = qkv.shape
    D = head_dim
    Hq, Hk, Hv = num_heads_q, num_heads_k, num_heads_v
    if (D == 128 and qkv.dtype in (torch.bfloat16, torch.float16)
            and not output_fp8):
        # fused CDNA4 kernel (csrc/qk_rope.hip): one wave per (token, head)
        from ._lib import get_ext

        cos, sin = _rope3d_cache(ppf, pph, ppw, D, base, str(qkv.device))
        q = q_out if q_out is not None else torch.empty(
            B, S, Hq, D, dtype=qkv.dtype, device=qkv.device)
        k = k_out if k_out is not None else torch.empty(
            B, S, Hk, D, dtype=qkv.dtype, device=qkv.device)
        v = v_out if v_out is not None else torch.empty(
            B, S, Hv, D, dtype=qkv.dtype, device=qkv.device)
        get_ext().qk_rope(qkv.contiguous(), q_weight.reshape(-1)[-D:].contiguous(),
                          k_weight.reshape(-1)[-D:].contiguous(),
                          cos.contiguous(), sin.contiguous(),
                          q.view(B * S, Hq, D), k.view(B * S, Hk, D),
                          v.view(B * S, Hv, D), S, Hq, Hk, Hv, D, eps,
                          attention_factor, is_qk_norm)
        return q, k, v
    q, k, v = qkv.split([Hq * D, Hk * D, Hv * D], dim=-1)
    q = q.reshape(B, S, Hq, D)
    k = k.reshape(B, S, Hk, D)
    v = v.reshape(B, S, Hv, D)
    if is_qk_norm:
        qf = q.float()
        q = (qf * torch.rsqrt(qf.pow(2).mean(-1, keepdim=True) + eps)
             * q_weight.float().view(-1)[-D:]).to(qkv.dtype)
        kf = k.float()
        k = (kf * torch.rsqrt(kf.pow(2).mean(-1, keepdim=True) + eps)
             * k_weight.float().view(-1)[-D:]).to(qkv.dtype)
    cos, sin = _rope3d_cache(ppf, pph, ppw, D, base, str(qkv.device))

    def rope(x):
        x1, x2 = x.float().unflatten(-1, (-1, 2)).unbind(-1)
        c = cos[None, :, None, 0::2]
        s = sin[None, :, None, 1::2]
        out = torch.empty_like(x, dtype=torch.float32)
        out[..., 0::2] = x1 * c - x2 * s
        out[..., 1::2] = x1 * s + x2 * c
        return (out * attention_factor).to(x.dtype)

    q, k = rope(q), rope(k)
    if output_fp8:
        q = (q.float() / output_quant_scale).to(torch.float8_e4m3fn)
        k = (k.float() / output_quant_scale).to(torch.float8_e4m3fn)
        v = (v.float() / v_quant_scale).to(torch.float8_e4m3fn)
    for dst, src in ((q_out, q), (k_out, k), (v_out, v)):
        if dst is not None:
            dst.copy_(src)
    return q, k, v


def _dit_tail(res_f32, out_dtype, residual_out, norm_out, gamma=None,
              beta=None, scale=None, shift=None, eps=1e-6):
    H = res_f32.shape[-1]
    n = torch.nn.functional.layer_norm(
        res_f32, [H],
        gamma.float() if gamma is not None else None,
        beta.float() if beta is not None else None, eps)
    if scale is not None:
        n = n * (1 + scale.float()) + shift.float()
    r = res_f32.to(out_dtype)
    n = n.to(out_dtype)
    if residual_out is not None:
        residual_out.copy_(r)
        r = residual_out
    if norm_out is not None:
        norm_out.copy_(n)
        n = norm_out
    return r, n


def fused_dit_gate_residual_layernorm_gamma_beta(
    input: torch.Tensor, residual: torch.Tensor, gate: torch.Tensor,
    gamma: torch.Tensor, beta: torch.Tensor, *, gate_bias=None,
    epsilon: float = 1e-6, input_global_scaling_factor=None,
    residual_out=None, norm_out=None, sf_out=None,
):
    r"""DiT adaLN epilogue (reference flashinfer/norm/__init__.py
    fused_dit_gate_residual_layernorm_gamma_beta:1138):
    ``residual += input * gate; norm = LN(residual, gamma, beta)``."""
    g = gate.float() + (gate_bias.float() if gate_bias is not None else 0.0)
    res = residual.float() + input.float() * g
    return _dit_tail(res, input.dtype, residual_out, norm_out,
                     gamma=gamma, beta=beta, eps=epsilon)


def fused_dit_gate_residual_layernorm_scale_shift(
    input: torch.Tensor, residual: torch.Tensor, gate: torch.Tensor,
    scale: torch.Tensor, shift: torch.Tensor, *, gate_bias=None,
    scale_bias=None, shift_bias=None, epsilon: float = 1e-6,
    input_global_scaling_factor=None, residual_out=None, norm_out=None,
    sf_out=None,
):
    r"""``residual += input * gate; norm = LN(residual) * (1+scale) + shift``
    (reference :1267)."""
    g = gate.float() + (gate_bias.float() if gate_bias is not None else 0.0)
    sc = scale.float() + (scale_bias.float() if scale_bias is not None else 0.0)
    sh = shift.float() + (shift_bias.float() if shift_bias is not None else 0.0)
    res = residual.float() + input.float() * g
    return _dit_tail(res, input.dtype, residual_out, norm_out,
                     scale=sc, shift=sh, eps=epsilon)


def fused_dit_residual_layernorm_scale_shift(
    input: torch.Tensor, scale: torch.Tensor, shift: torch.Tensor,
    residual: Optional[torch.Tensor] = None, *, scale_bias=None,
    shift_bias=None, epsilon: float = 1e-6,
    input_global_scaling_factor=None, residual_out=None, norm_out=None,
    sf_out=None,
):
    r"""``residual += input; norm = LN(residual) * (1+scale) + shift``
    (reference :1389)."""
    sc = scale.float() + (scale_bias.float() if scale_bias is not None else 0.0)
    sh = shift.float() + (shift_bias.float() if shift_bias is not None else 0.0)
    res = input.float() if residual is None else residual.float() + input.float()
    return _dit_tail(res, input.dtype, residual_out, norm_out,
                     scale=sc, shift=sh, eps=epsilon)
