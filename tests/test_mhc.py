"""GPU numerics: mHC hyper-connection ops + concat_mla_k vs fp32 references
(math per reference tests/mhc/test_mhc_pre_big_fuse.py, test_mhc_post.py)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

RMS_EPS = 1e-6
PRE_EPS = 1e-6
SINK_EPS = 1e-6
REPEAT = 20


def _sinkhorn(x, repeat=REPEAT, eps=SINK_EPS):
    x = x.softmax(dim=-1) + eps
    x = x / (x.sum(dim=-2, keepdim=True) + eps)
    for _ in range(repeat - 1):
        x = x / (x.sum(dim=-1, keepdim=True) + eps)
        x = x / (x.sum(dim=-2, keepdim=True) + eps)
    return x


def _pre_ref(dot_mix, sqrsum, residual, scale, base, k):
    hc = residual.shape[-2]
    rstd = torch.rsqrt(sqrsum.float().unsqueeze(-1) / float(k) + RMS_EPS)
    mixes = dot_mix.float() * rstd
    pre = torch.sigmoid(mixes[..., :hc] * scale[0] + base[:hc]).unsqueeze(-1) + PRE_EPS
    post = torch.sigmoid(mixes[..., hc:2*hc] * scale[1] + base[hc:2*hc]).unsqueeze(-1)
    comb = _sinkhorn((mixes[..., 2*hc:] * scale[2] + base[2*hc:]).view(
        *residual.shape[:-2], hc, hc))
    layer_input = (pre * residual.float()).sum(dim=-2)
    return post, comb, layer_input


def test_mhc_pre_big_fuse():
    from flashinfer_amd.mhc import mhc_pre_big_fuse

    torch.manual_seed(42)
    T, H = 6, 4096
    residual = (torch.randn(T, 4, H, device="cuda") * 0.01).bfloat16()
    scale = torch.randn(3, device="cuda") * 0.1
    base = torch.randn(24, device="cuda") * 0.1
    dot_mix = torch.randn(T, 24, device="cuda")
    sqrsum = residual.float().square().sum(dim=(-1, -2))
    k = 4 * H
    post, comb, li = mhc_pre_big_fuse(dot_mix, sqrsum, residual, scale, base, k)
    rp, rc, rl = _pre_ref(dot_mix, sqrsum, residual, scale, base, k)
    torch.testing.assert_close(post, rp, atol=2e-3, rtol=2e-3)
    torch.testing.assert_close(comb, rc, atol=2e-3, rtol=2e-3)
    torch.testing.assert_close(li.float(), rl, atol=1e-2, rtol=1e-2)


def test_mhc_pre_big_fuse_with_prenorm():
    from flashinfer_amd.mhc import mhc_pre_big_fuse_with_prenorm

    torch.manual_seed(1)
    T, H = 4, 1024
    residual = (torch.randn(T, 4, H, device="cuda") * 0.01).bfloat16()
    scale = torch.randn(3, device="cuda") * 0.1
    base = torch.randn(24, device="cuda") * 0.1
    dot_mix = torch.randn(T, 24, device="cuda")
    post, comb, li = mhc_pre_big_fuse_with_prenorm(dot_mix, residual, scale, base)
    sqrsum = residual.flatten(-2).float().square().sum(-1)
    rp, rc, rl = _pre_ref(dot_mix, sqrsum, residual, scale, base, 4 * H)
    torch.testing.assert_close(post, rp, atol=2e-3, rtol=2e-3)
    torch.testing.assert_close(comb, rc, atol=2e-3, rtol=2e-3)
    torch.testing.assert_close(li.float(), rl, atol=1e-2, rtol=1e-2)


def test_mhc_post():
    from flashinfer_amd.mhc import mhc_post

    torch.manual_seed(2)
    T, H = 5, 2048
    x = torch.randn(T, H, device="cuda").bfloat16()
    residual = torch.randn(T, 4, H, device="cuda").bfloat16()
    post = torch.rand(T, 4, device="cuda")
    comb = torch.rand(T, 4, 4, device="cuda")
    out = mhc_post(x, residual, post, comb)
    ref = (x.float()[:, None] * post[..., None]
           + torch.einsum("toh,ton->tnh", residual.float(), comb))
    torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float8_e4m3fn])
def test_concat_mla_k(dtype):
    from flashinfer_amd.concat_ops import concat_mla_k

    torch.manual_seed(3)
    T, Hk, nope, rope = 64, 128, 128, 64
    k_nope = torch.randn(T, Hk, nope, device="cuda").to(dtype)
    k_rope = torch.randn(T, 1, rope, device="cuda").to(dtype)
    k = torch.empty(T, Hk, nope + rope, dtype=dtype, device="cuda")
    concat_mla_k(k, k_nope, k_rope)
    ref = torch.cat([k_nope.float(),
                     k_rope.float().expand(T, Hk, rope)], dim=-1)
    torch.testing.assert_close(k.float(), ref)
