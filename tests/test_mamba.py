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


def test_mamba_chunk_scan_combined():
    from flashinfer_amd.mamba import mamba_chunk_scan_combined

    torch.manual_seed(0)
    Bn, L, H, G, P, N = 2, 100, 4, 2, 64, 128
    x = (torch.randn(Bn, L, H, P, device="cuda") / 4).bfloat16()
    dt = torch.rand(Bn, L, H, device="cuda") * 0.5
    A = -torch.rand(H, device="cuda").float()
    Bm = torch.randn(Bn, L, G, N, device="cuda").bfloat16() / 4
    Cm = torch.randn(Bn, L, G, N, device="cuda").bfloat16() / 4
    D = torch.rand(H, device="cuda").float()
    z = torch.randn(Bn, L, H, P, device="cuda").bfloat16()
    dt_bias = torch.rand(H, device="cuda").float()
    S0 = torch.randn(Bn, H, P, N, device="cuda") / 8

    out, Sf = mamba_chunk_scan_combined(
        x, dt, A, Bm, Cm, 128, D=D, z=z, dt_bias=dt_bias, dt_softplus=True,
        initial_states=S0, return_final_states=True)

    # sequential fp32 reference
    S = S0.clone().float()
    ref = torch.zeros(Bn, L, H, P, device="cuda")
    rep = H // G
    for t in range(L):
        d = torch.nn.functional.softplus(dt[:, t] + dt_bias)      # [B, H]
        dA = torch.exp(d * A)                                      # [B, H]
        Bt = Bm[:, t].float().repeat_interleave(rep, 1)            # [B, H, N]
        Ct = Cm[:, t].float().repeat_interleave(rep, 1)
        S = S * dA[..., None, None] + torch.einsum(
            "bhp,bhn->bhpn", d[..., None] * x[:, t].float(), Bt)
        y = torch.einsum("bhpn,bhn->bhp", S, Ct) + D[None, :, None] * x[:, t].float()
        zt = z[:, t].float()
        ref[:, t] = y * zt * torch.sigmoid(zt)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(Sf, S, atol=5e-2, rtol=5e-2)


def test_mamba_chunk_scan_matches_ssu_steps():
    """Prefill scan must agree with stepping selective_state_update per token."""
    from flashinfer_amd.mamba import mamba_chunk_scan_combined, selective_state_update

    torch.manual_seed(1)
    Bn, L, H, G, P, N = 2, 17, 2, 2, 64, 64
    x = (torch.randn(Bn, L, H, P, device="cuda") / 4).bfloat16()
    dt = (torch.rand(Bn, L, H, device="cuda") * 0.5).bfloat16()
    A = -torch.rand(H, device="cuda").float()
    Bm = torch.randn(Bn, L, G, N, device="cuda").bfloat16() / 4
    Cm = torch.randn(Bn, L, G, N, device="cuda").bfloat16() / 4

    out, Sf = mamba_chunk_scan_combined(x, dt.float(), A, Bm, Cm,
                                        return_final_states=True)
    S = torch.zeros(Bn, H, P, N, device="cuda")
    for t in range(L):
        y = selective_state_update(
            S, x[:, t], dt[:, t], A.bfloat16(), Bm[:, t], Cm[:, t], None)
        torch.testing.assert_close(out[:, t].float(), y.float(),
                                   atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(Sf, S, atol=3e-2, rtol=3e-2)
