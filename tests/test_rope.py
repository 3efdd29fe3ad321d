"""GPU numerics: RoPE kernels vs fp32 PyTorch reference."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def rope_ref(x, pos, rot_dim, theta=1e4, interleave=False, scale=1.0):
    # x [nnz, H, D] -> rotated fp32
    xf = x.float()
    nnz, H, D = xf.shape
    half = rot_dim // 2
    j = torch.arange(half, device=x.device, dtype=torch.float32)
    freq = theta ** (-2 * j / rot_dim) / scale
    ang = pos.float()[:, None] * freq[None, :]  # [nnz, half]
    cos, sin = torch.cos(ang)[:, None, :], torch.sin(ang)[:, None, :]
    out = xf.clone()
    if interleave:
        x1, x2 = xf[..., 0:rot_dim:2], xf[..., 1:rot_dim:2]
        out[..., 0:rot_dim:2] = x1 * cos - x2 * sin
        out[..., 1:rot_dim:2] = x2 * cos + x1 * sin
    else:
        x1, x2 = xf[..., :half], xf[..., half:rot_dim]
        out[..., :half] = x1 * cos - x2 * sin
        out[..., half:rot_dim] = x2 * cos + x1 * sin
    return out


@pytest.mark.parametrize("interleave", [False, True])
@pytest.mark.parametrize("D,rot", [(128, 128), (128, 64), (64, 64)])
def test_apply_rope_pos_ids(interleave, D, rot):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    nnz, Hq, Hkv = 123, 8, 2
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, 4096, (nnz,), dtype=torch.int32, device="cuda")
    q_out, k_out = fi.apply_rope_pos_ids(
        q, k, pos, rotary_dim=rot, interleave=interleave
    )
    torch.testing.assert_close(
        q_out.float(), rope_ref(q, pos, rot, interleave=interleave),
        atol=2e-2, rtol=2e-2,
    )
    torch.testing.assert_close(
        k_out.float(), rope_ref(k, pos, rot, interleave=interleave),
        atol=2e-2, rtol=2e-2,
    )


def test_apply_rope_indptr_offsets():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    lens = [5, 1, 17]
    offsets = torch.tensor([100, 0, 2000], dtype=torch.int32, device="cuda")
    indptr = torch.tensor([0, 5, 6, 23], dtype=torch.int32, device="cuda")
    nnz = 23
    q = torch.randn(nnz, 4, 64, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, 2, 64, dtype=torch.bfloat16, device="cuda")
    q_out, k_out = fi.apply_rope(q, k, indptr, offsets)
    pos = torch.cat(
        [offsets[i] + torch.arange(L, device="cuda") for i, L in enumerate(lens)]
    )
    torch.testing.assert_close(
        q_out.float(), rope_ref(q, pos, 64), atol=2e-2, rtol=2e-2
    )


def test_rope_cos_sin_cache():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    D, max_pos = 128, 8192
    half = D // 2
    j = torch.arange(half, device="cuda", dtype=torch.float32)
    inv_freq = 1e4 ** (-2 * j / D)
    t = torch.arange(max_pos, device="cuda", dtype=torch.float32)
    ang = t[:, None] * inv_freq[None, :]
    cache = torch.cat([torch.cos(ang), torch.sin(ang)], dim=-1)

    nnz, Hq, Hkv = 77, 8, 2
    q = torch.randn(nnz, Hq * D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, Hkv * D, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, max_pos, (nnz,), dtype=torch.int32, device="cuda")
    q_out, k_out = fi.apply_rope_with_cos_sin_cache(pos, q, k, D, cache, is_neox=True)
    ref = rope_ref(q.view(nnz, Hq, D), pos, D)
    torch.testing.assert_close(
        q_out.view(nnz, Hq, D).float(), ref, atol=2e-2, rtol=2e-2
    )


def test_llama31_rope_matches_hf_formula():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    D = 128
    nnz = 64
    factor, low, high, old_ctx, theta = 8.0, 1.0, 4.0, 8192, 5e5
    q = torch.randn(nnz, 4, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, 1, D, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, 32768, (nnz,), dtype=torch.int32, device="cuda")
    q_out, _ = fi.apply_llama31_rope_pos_ids(
        q, k, pos, rope_scale=factor, rope_theta=theta,
        low_freq_factor=low, high_freq_factor=high, old_context_len=old_ctx,
    )
    # HF reference
    j = torch.arange(D // 2, device="cuda", dtype=torch.float32)
    inv_freq = theta ** (-2 * j / D)
    wavelen = 2 * math.pi / inv_freq
    smooth = (old_ctx / wavelen - low) / (high - low)
    smooth = smooth.clamp(0, 1)
    inv_freq_sc = (1 - smooth) * inv_freq / factor + smooth * inv_freq
    ang = pos.float()[:, None] * inv_freq_sc[None, :]
    cos, sin = torch.cos(ang)[:, None, :], torch.sin(ang)[:, None, :]
    xf = q.float()
    half = D // 2
    ref = xf.clone()
    ref[..., :half] = xf[..., :half] * cos - xf[..., half:] * sin
    ref[..., half:] = xf[..., half:] * cos + xf[..., :half] * sin
    torch.testing.assert_close(q_out.float(), ref, atol=2e-2, rtol=2e-2)


def test_inplace_rope_variants_match_out_of_place():
    """The inplace/pos-ids/llama-3.1 export variants share the same kernel —
    each must match its out-of-place sibling exactly."""
    import flashinfer_amd as fi

    torch.manual_seed(3)
    nnz, Hq, Hkv, D = 64, 8, 2, 128
    indptr = torch.tensor([0, 40, nnz], dtype=torch.int32, device="cuda")
    offsets = torch.tensor([5, 17], dtype=torch.int32, device="cuda")

    def fresh():
        g = torch.Generator("cuda").manual_seed(11)
        q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda",
                        generator=g)
        k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda",
                        generator=g)
        return q, k

    # plain rope: inplace == out-of-place
    q, k = fresh()
    qo, ko = fi.apply_rope(q, k, indptr, offsets)
    qi, ki = fresh()
    fi.apply_rope_inplace(qi, ki, indptr, offsets)
    assert torch.equal(qo, qi) and torch.equal(ko, ki)

    # pos-ids inplace == pos-ids out-of-place
    pos_ids = torch.arange(nnz, dtype=torch.int32, device="cuda") % 40
    q, k = fresh()
    qo, ko = fi.apply_rope_pos_ids(q, k, pos_ids)
    qi, ki = fresh()
    fi.apply_rope_pos_ids_inplace(qi, ki, pos_ids)
    assert torch.equal(qo, qi) and torch.equal(ko, ki)

    # llama-3.1 wavelength-dependent scaling: inplace == out-of-place,
    # and differs from plain rope (the frequency remap must be active)
    q, k = fresh()
    qo, ko = fi.apply_llama31_rope(q, k, indptr, offsets)
    qi, ki = fresh()
    fi.apply_llama31_rope_inplace(qi, ki, indptr, offsets)
    assert torch.equal(qo, qi) and torch.equal(ko, ki)
    qp, _ = fresh()
    qplain, _ = fi.apply_rope(qp, k.clone(), indptr, offsets)
    assert not torch.equal(qo, qplain)
    qi, ki = fresh()
    fi.apply_llama31_rope_pos_ids_inplace(
        qi, ki, torch.cat([torch.arange(40, device="cuda"),
                           torch.arange(nnz - 40, device="cuda")]).int() +
        torch.cat([torch.full((40,), 5, device="cuda"),
                   torch.full((nnz - 40,), 17, device="cuda")]).int())
    assert torch.equal(qi, qo) and torch.equal(ki, ko)


def test_rope_cos_sin_cache_inplace():
    import flashinfer_amd as fi

    torch.manual_seed(4)
    nnz, H, D = 32, 4, 64
    pos_ids = torch.arange(nnz, dtype=torch.int32, device="cuda")
    inv = 1.0 / (1e4 ** (torch.arange(0, D // 2, device="cuda").float()
                         / (D // 2)))
    ang = pos_ids.float()[:, None] * inv[None, :]
    cos_sin = torch.cat([ang.cos(), ang.sin()], -1)
    q3 = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    k3 = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    # vLLM convention: FLAT [tokens, heads*head_size]
    q = q3.reshape(nnz, H * D).clone()
    k = k3.reshape(nnz, H * D).clone()
    fi.apply_rope_with_cos_sin_cache_inplace(pos_ids, q, k, D, cos_sin,
                                             is_neox=True)
    ref_q, ref_k = fi.apply_rope_pos_ids(q3, k3, pos_ids)
    torch.testing.assert_close(q.view(nnz, H, D).float(), ref_q.float(),
                               atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(k.view(nnz, H, D).float(), ref_k.float(),
                               atol=5e-2, rtol=5e-2)
