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


def _ref_chunk(q, k, v, g, beta, cu, scale, S0=None):
    """Sequential f32 reference of the gated delta rule over ragged seqs."""
    T, H, D = q.shape
    ns = cu.numel() - 1
    S = (S0.clone().float() if S0 is not None
         else torch.zeros(ns, H, D, D, device=q.device))
    o = torch.zeros(T, H, D, device=q.device)
    for s in range(ns):
        for t in range(int(cu[s]), int(cu[s + 1])):
            a = g[t].float()  # [H] or [H, D]
            Sf = S[s] * (a[..., None] if a.dim() == 2 else a[:, None, None])
            kv = torch.einsum("hi,hij->hj", k[t].float(), Sf)
            delta = beta[t].float()[:, None] * (v[t].float() - kv)
            S[s] = Sf + torch.einsum("hi,hj->hij", k[t].float(), delta)
            o[t] = scale * torch.einsum("hi,hij->hj", q[t].float(), S[s])
    return o, S


def test_chunk_gated_delta_rule_ragged():
    from flashinfer_amd.gdn import chunk_gated_delta_rule

    torch.manual_seed(0)
    H, D = 4, 128
    lens = [37, 0, 100, 5]
    cu = torch.tensor([0] + list(torch.cumsum(torch.tensor(lens), 0)),
                      dtype=torch.int32, device="cuda")
    T = sum(lens)
    q = torch.randn(T, H, D, device="cuda").bfloat16()
    k = torch.nn.functional.normalize(
        torch.randn(T, H, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(T, H, D, device="cuda") / 4).bfloat16()
    g = torch.rand(T, H, device="cuda") * 0.8 + 0.1
    beta = torch.rand(T, H, device="cuda")
    S0 = torch.randn(len(lens), H, D, D, device="cuda") / 8
    scale = D ** -0.5

    o, Sf = chunk_gated_delta_rule(q, k, v, g, beta, cu, scale,
                                   initial_state=S0, output_final_state=True)
    o_ref, S_ref = _ref_chunk(q, k, v, g, beta, cu, scale, S0)
    torch.testing.assert_close(o.float(), o_ref, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(Sf, S_ref, atol=5e-2, rtol=5e-2)
    # zero-length sequence: state must pass through untouched
    torch.testing.assert_close(Sf[1], S0[1].float())


def test_chunk_kda_per_channel():
    from flashinfer_amd.gdn import chunk_kda

    torch.manual_seed(1)
    H, D = 2, 64
    lens = [65, 31]
    cu = torch.tensor([0] + list(torch.cumsum(torch.tensor(lens), 0)),
                      dtype=torch.int32, device="cuda")
    T = sum(lens)
    q = torch.randn(T, H, D, device="cuda").bfloat16()
    k = torch.nn.functional.normalize(
        torch.randn(T, H, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(T, H, D, device="cuda") / 4).bfloat16()
    g = torch.rand(T, H, D, device="cuda") * 0.8 + 0.1
    beta = torch.rand(T, H, device="cuda")

    o, Sf = chunk_kda(q, k, v, g, beta, cu, output_final_state=True)
    o_ref, S_ref = _ref_chunk(q, k, v, g, beta, cu, D ** -0.5)
    torch.testing.assert_close(o.float(), o_ref, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(Sf, S_ref, atol=5e-2, rtol=5e-2)


def test_chunk_matches_decode_steps():
    """Prefill over a 1-token-at-a-time loop of the decode kernel must agree."""
    from flashinfer_amd.gdn import chunk_gated_delta_rule, gdn_fused_decode_step

    torch.manual_seed(2)
    H, D, T = 4, 128, 24
    cu = torch.tensor([0, T], dtype=torch.int32, device="cuda")
    q = torch.randn(T, H, D, device="cuda").bfloat16()
    k = torch.nn.functional.normalize(
        torch.randn(T, H, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(T, H, D, device="cuda") / 4).bfloat16()
    g = torch.rand(T, H, device="cuda") * 0.8 + 0.1
    beta = torch.rand(T, H, device="cuda")

    o, Sf = chunk_gated_delta_rule(q, k, v, g, beta, cu, 1.0,
                                   output_final_state=True)
    S = torch.zeros(1, H, D, D, device="cuda")
    for t in range(T):
        od = gdn_fused_decode_step(S, q[t:t+1], k[t:t+1], v[t:t+1],
                                   g[t:t+1], beta[t:t+1])
        torch.testing.assert_close(o[t:t+1].float(), od.float(),
                                   atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(Sf, S, atol=3e-2, rtol=3e-2)
