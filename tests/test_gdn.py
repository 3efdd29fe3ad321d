"""GPU numerics: GDN decode step vs torch reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("state_f32", [True, False])
def test_gdn_decode_step(state_f32):
    from flashinfer_amd.gdn import gdn_fused_decode_step

    torch.manual_seed(0)
    B, H, Dk, Dv = 4, 8, 128, 128
    dtype = torch.bfloat16
    sdtype = torch.float32 if state_f32 else dtype
    S = (torch.randn(B, H, Dk, Dv, device="cuda") / 8).to(sdtype)
    S0 = S.clone()
    q = torch.nn.functional.normalize(torch.randn(B, H, Dk, device="cuda"), dim=-1).to(dtype)
    k = torch.nn.functional.normalize(torch.randn(B, H, Dk, device="cuda"), dim=-1).to(dtype)
    v = (torch.randn(B, H, Dv, device="cuda") / 4).to(dtype)
    g = torch.rand(B, H, device="cuda") * 0.9
    beta = torch.rand(B, H, device="cuda")
    o = gdn_fused_decode_step(S, q, k, v, g, beta)
    # reference
    Sf = S0.float() * g[..., None, None]
    kv = torch.einsum("bhi,bhij->bhj", k.float(), Sf)
    delta = beta[..., None] * (v.float() - kv)
    S_new = Sf + torch.einsum("bhi,bhj->bhij", k.float(), delta)
    o_ref = torch.einsum("bhi,bhij->bhj", q.float(), S_new)
    torch.testing.assert_close(S.float(), S_new, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(o.float(), o_ref, atol=5e-2, rtol=5e-2)
    # multi-step stability: run 8 steps, state stays finite
    for _ in range(8):
        gdn_fused_decode_step(S, q, k, v, g, beta)
    assert S.float().isfinite().all()


def test_kda_per_channel_gate():
    from flashinfer_amd.gdn import fused_kda_decode

    torch.manual_seed(0)
    B, H, Dk, Dv = 2, 4, 64, 128
    S = torch.randn(B, H, Dk, Dv, device="cuda") / 8
    S0 = S.clone()
    q = torch.randn(B, H, Dk, device="cuda").bfloat16()
    k = torch.randn(B, H, Dk, device="cuda").bfloat16()
    v = (torch.randn(B, H, Dv, device="cuda") / 4).bfloat16()
    g = torch.rand(B, H, Dk, device="cuda") * 0.9
    beta = torch.rand(B, H, device="cuda")
    o = fused_kda_decode(S, q, k, v, g, beta)
    Sf = S0 * g[..., None]
    kv = torch.einsum("bhi,bhij->bhj", k.float(), Sf)
    delta = beta[..., None] * (v.float() - kv)
    S_new = Sf + torch.einsum("bhi,bhj->bhij", k.float(), delta)
    o_ref = torch.einsum("bhi,bhij->bhj", q.float(), S_new)
    torch.testing.assert_close(S, S_new, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(o.float(), o_ref, atol=5e-2, rtol=5e-2)
