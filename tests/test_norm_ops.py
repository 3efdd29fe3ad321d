"""GPU numerics: norm + activation kernels vs fp32 PyTorch references."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _rmsnorm_ref(x, w, eps, wbias=0.0):
    xf = x.float()
    rms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return xf * rms * (w.float() + wbias)


@pytest.mark.parametrize("rows", [1, 17, 999])
@pytest.mark.parametrize("d", [128, 4096, 4100])
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
def test_rmsnorm(rows, d, dtype):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(rows, d, dtype=dtype, device="cuda")
    w = torch.randn(d, dtype=dtype, device="cuda")
    out = fi.rmsnorm(x, w, eps=1e-6)
    ref = _rmsnorm_ref(x, w, 1e-6)
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


def test_gemma_rmsnorm():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(64, 2048, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(2048, dtype=torch.bfloat16, device="cuda")
    out = fi.gemma_rmsnorm(x, w, eps=1e-6)
    ref = _rmsnorm_ref(x, w, 1e-6, wbias=1.0)
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16])
def test_fused_add_rmsnorm(dtype):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(33, 1024, dtype=dtype, device="cuda")
    r = torch.randn(33, 1024, dtype=dtype, device="cuda")
    w = torch.randn(1024, dtype=dtype, device="cuda")
    x2, r2 = x.clone(), r.clone()
    fi.fused_add_rmsnorm(x2, r2, w, eps=1e-6)
    ref_res = (x.float() + r.float()).to(dtype)
    ref_out = _rmsnorm_ref(ref_res, w, 1e-6)
    torch.testing.assert_close(r2.float(), ref_res.float(), atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(x2.float(), ref_out, atol=2e-2, rtol=2e-2)


def test_layernorm():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(65, 768, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(768, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(768, dtype=torch.bfloat16, device="cuda")
    out = fi.layernorm(x, w, b, eps=1e-5)
    ref = torch.nn.functional.layer_norm(
        x.float(), (768,), w.float(), b.float(), 1e-5
    )
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("act", ["silu", "gelu", "gelu_tanh"])
@pytest.mark.parametrize("shape", [(7, 256), (1025, 2816 * 2 // 2)])
def test_act_and_mul(act, shape):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    rows, d = shape
    x = torch.randn(rows, 2 * d, dtype=torch.bfloat16, device="cuda")
    fn = {
        "silu": fi.silu_and_mul,
        "gelu": fi.gelu_and_mul,
        "gelu_tanh": fi.gelu_tanh_and_mul,
    }[act]
    out = fn(x)
    g, u = x.float().chunk(2, dim=-1)
    ref = {
        "silu": torch.nn.functional.silu(g),
        "gelu": torch.nn.functional.gelu(g),
        "gelu_tanh": torch.nn.functional.gelu(g, approximate="tanh"),
    }[act] * u
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


def test_rmsnorm_quant():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(64, 4096, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
    scale = torch.tensor([0.5], dtype=torch.float32, device="cuda")
    out = torch.empty(64, 4096, dtype=torch.float8_e4m3fn, device="cuda")
    fi.rmsnorm_quant(out, x, w, scale)
    xf = x.float()
    ref = xf / torch.sqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    ref = (ref / scale).to(torch.float8_e4m3fn).float()
    torch.testing.assert_close(out.float(), ref, atol=0.2, rtol=0.2)


def test_fused_add_rmsnorm_quant():
    import flashinfer_amd as fi

    torch.manual_seed(1)
    x = torch.randn(32, 2048, dtype=torch.bfloat16, device="cuda")
    r = torch.randn(32, 2048, dtype=torch.bfloat16, device="cuda")
    r0 = r.clone()
    w = torch.randn(2048, dtype=torch.bfloat16, device="cuda")
    out = torch.empty(32, 2048, dtype=torch.float8_e4m3fn, device="cuda")
    fi.fused_add_rmsnorm_quant(out, x, r, w, 2.0)
    rf = (x.float() + r0.float())
    torch.testing.assert_close(r.float(), rf, atol=2e-2, rtol=2e-2)
    rf = r.float()  # quantize path uses the bf16-rounded residual
    ref = rf / torch.sqrt(rf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    ref = (ref / 2.0).to(torch.float8_e4m3fn).float()
    torch.testing.assert_close(out.float(), ref, atol=0.2, rtol=0.2)


def test_fused_rmsnorm_silu():
    import flashinfer_amd as fi

    torch.manual_seed(2)
    x = torch.randn(16, 1024, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(1024, dtype=torch.bfloat16, device="cuda")
    out = fi.fused_rmsnorm_silu(x, w)
    xf = x.float()
    ref = xf / torch.sqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    ref = ref * torch.sigmoid(ref)
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


def test_layernorm_quant():
    import flashinfer_amd as fi

    torch.manual_seed(3)
    x = torch.randn(32, 1024, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(1024, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(1024, dtype=torch.bfloat16, device="cuda")
    out = torch.empty(32, 1024, dtype=torch.float8_e4m3fn, device="cuda")
    fi.layernorm_quant(out, x, w, 0.5, bias=b)
    xf = x.float()
    ref = (xf - xf.mean(-1, keepdim=True)) / torch.sqrt(
        xf.var(-1, unbiased=False, keepdim=True) + 1e-6) * w.float() + b.float()
    ref = (ref / 0.5).to(torch.float8_e4m3fn).float()
    torch.testing.assert_close(out.float(), ref, atol=0.25, rtol=0.25)


def test_fused_qk_rmsnorm_rope_3d():
    """WAN-style qk-norm + interleaved 3D rope vs the documented reference
    math (reference tests/norm/test_fused_qk_rmsnorm_rope.py)."""
    import flashinfer_amd as fi

    torch.manual_seed(5)
    B, ppf, pph, ppw = 2, 3, 4, 5
    S = ppf * pph * ppw
    H, D = 4, 128
    qkv = torch.randn(B, S, 3 * H * D, dtype=torch.bfloat16, device="cuda")
    qw = torch.randn(D, dtype=torch.bfloat16, device="cuda")
    kw = torch.randn(D, dtype=torch.bfloat16, device="cuda")
    q, k, v = fi.fused_qk_rmsnorm_rope(
        qkv, qw, kw, ppf=ppf, pph=pph, ppw=ppw,
        num_heads_q=H, num_heads_k=H, num_heads_v=H, head_dim=D)
    assert q.shape == (B, S, H, D)

    # reference math
    h_dim = w_dim = 2 * (D // 6)
    t_dim = D - h_dim - w_dim

    def table(dim, length):
        inv = 1.0 / (10000.0 ** (torch.arange(0, dim, 2, device="cuda",
                                              dtype=torch.float64) / dim))
        pos = torch.arange(length, device="cuda", dtype=torch.float64)
        fr = torch.einsum("i,j->ij", pos, inv)
        return (torch.repeat_interleave(torch.cos(fr), 2, -1),
                torch.repeat_interleave(torch.sin(fr), 2, -1))

    mx = max(ppf, pph, ppw)
    tc, ts = table(t_dim, mx)
    hc, hs = table(h_dim, mx)
    wc, ws = table(w_dim, mx)
    tok = torch.arange(S, device="cuda")
    pt, ph, pw = tok // (pph * ppw), (tok // ppw) % pph, tok % ppw
    cos = torch.cat([tc[pt], hc[ph], wc[pw]], -1).float()
    sin = torch.cat([ts[pt], hs[ph], ws[pw]], -1).float()

    qkv3 = qkv.view(B, S, 3, H, D)
    qn = qkv3[:, :, 0].float()
    qn = qn * torch.rsqrt(qn.pow(2).mean(-1, keepdim=True) + 1e-6) * qw.float()
    qn = qn.to(torch.bfloat16).float()
    x1, x2 = qn.unflatten(-1, (-1, 2)).unbind(-1)
    ref_q = torch.empty_like(qn)
    ref_q[..., 0::2] = x1 * cos[None, :, None, 0::2] - x2 * sin[None, :, None, 1::2]
    ref_q[..., 1::2] = x1 * sin[None, :, None, 1::2] + x2 * cos[None, :, None, 0::2]
    torch.testing.assert_close(q.float(), ref_q, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(v.float(), qkv3[:, :, 2].float())


def test_gemma_fused_add_rmsnorm():
    """Gemma variant: norm weight is (1 + w) — export previously untested."""
    import flashinfer_amd as fi

    torch.manual_seed(5)
    x = torch.randn(32, 512, dtype=torch.bfloat16, device="cuda")
    res = torch.randn(32, 512, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(512, dtype=torch.bfloat16, device="cuda") * 0.1
    x2, r2 = x.clone(), res.clone()
    fi.gemma_fused_add_rmsnorm(x2, r2, w)
    s = x.float() + res.float()
    ref = (s * torch.rsqrt(s.pow(2).mean(-1, keepdim=True) + 1e-6)
           * (1.0 + w.float()))
    torch.testing.assert_close(r2.float(), s, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(x2.float(), ref, atol=3e-2, rtol=3e-2)
