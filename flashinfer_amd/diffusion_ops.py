"""Diffusion / DiT ops (reference parity: flashinfer/diffusion_ops/__init__.py
— re-exports of the fused DiT adaLN layernorms and the qk-norm+3D-rope
epilogue used by video diffusion models)."""
from .norm import (
    fused_dit_gate_residual_layernorm_gamma_beta,
    fused_dit_gate_residual_layernorm_scale_shift,
    fused_dit_residual_layernorm_scale_shift,
    fused_qk_rmsnorm_rope,
)

__all__ = [
    "fused_dit_gate_residual_layernorm_gamma_beta",
    "fused_dit_gate_residual_layernorm_scale_shift",
    "fused_dit_residual_layernorm_scale_shift",
    "fused_qk_rmsnorm_rope",
]
