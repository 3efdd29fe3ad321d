"""GPU numerics: selective_state_update vs einsum reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("state_f32", [True, False])
@pytest.mark.parametrize("with_z", [True, False])
def test_selective_state_update(state_f32, with_z):
    from flashinfer_amd.mamba import selective_state_update

    torch.manual_seed(0)
    B, H, P, S, G = 3, 8, 64, 128, 2
    dtype = torch.bfloat16
    sdtype = torch.float32 if state_f32 else dtype
    state = torch.randn(B, H, P, S, dtype=sdtype, device="cuda")
    state0 = state.clone()
    x = torch.randn(B, H, P, dtype=dtype, device="cuda")
    dt = torch.randn(B, H, dtype=dtype, device="cuda")
    A = -torch.rand(H, dtype=dtype, device="cuda")
    Bm = torch.randn(B, G, S, dtype=dtype, device="cuda")
    Cm = torch.randn(B, G, S, dtype=dtype, device="cuda")
    D = torch.randn(H, dtype=dtype, device="cuda")
    z = torch.randn(B, H, P, dtype=dtype, device="cuda") if with_z else None
    dt_bias = torch.randn(H, dtype=dtype, device="cuda")
    y = selective_state_update(state, x, dt, A, Bm, Cm, D=D, z=z,
                               dt_bias=dt_bias, dt_softplus=True)
    # reference
    dtf = torch.nn.functional.softplus(dt.float() + dt_bias.float())
    dA = torch.exp(dtf * A.float())  # [B, H]
    Bg = Bm.float().repeat_interleave(H // G, dim=1)  # [B, H, S]
    Cg = Cm.float().repeat_interleave(H // G, dim=1)
    ref_state = state0.float() * dA[..., None, None] + (
        dtf[..., None] * x.float()
    )[..., None] * Bg[:, :, None, :]
    ref_y = torch.einsum("bhps,bhs->bhp", ref_state, Cg) + D.float()[:, None] * x.float()
    if with_z:
        ref_y = ref_y * torch.nn.functional.silu(z.float())
    torch.testing.assert_close(state.float(), ref_state, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(y.float(), ref_y, atol=2e-1, rtol=5e-2)
