"""API-surface parity: aliases + composed compatibility entry points."""
import math

import pytest
import torch

import flashinfer_amd as fi


def test_top_level_exports_present():
    for name in [
        "autotune", "fast_decode_plan", "jit", "bmm_bf16",
        "BatchAttentionWithAttentionSinkWrapper",
        "BatchDecodeMlaWithPagedKVCacheWrapper",
        "BatchDecodeWithSharedPrefixPagedKVCacheWrapper",
        "BatchPrefillWithSharedPrefixPagedKVCacheWrapper",
        "recurrent_kda", "packed_kda_decode", "RecurrentKDAPrefillWrapper",
        "RoutingMethodType", "ActivationType", "is_gated_activation",
        "trtllm_bf16_moe", "trtllm_fp8_per_tensor_scale_moe",
        "next_positive_power_of_2", "gdn_fused_decode_step_supported",
        "append_paged_mla_kv_cache", "single_prefill_with_kv_cache_return_lse",
        "fused_dit_gate_residual_layernorm_gamma_beta",
    ]:
        assert hasattr(fi, name), name
    assert fi.next_positive_power_of_2(7) == 8
    assert fi.is_gated_activation("silu")
    assert fi.gdn_fused_decode_step_supported()


@pytest.mark.gpu
def test_shared_prefix_decode_wrapper():
    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    B, shared, uniq = 3, 128, 64
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    ks = torch.randn(shared, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vs = torch.randn(shared, Hkv, D, dtype=torch.bfloat16, device="cuda")
    np_ = B * (uniq // page)
    kc = torch.randn(np_, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(np_, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    indptr = torch.arange(0, np_ + 1, uniq // page, dtype=torch.int32,
                          device="cuda")
    indices = torch.arange(np_, dtype=torch.int32, device="cuda")
    lpl = torch.full((B,), page, dtype=torch.int32, device="cuda")
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithSharedPrefixPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page)
    out = w.forward(q, ks, vs, (kc, vc))
    # dense reference
    G = Hq // Hkv
    for b in range(B):
        kv = torch.cat([ks, kc.view(-1, Hkv, D)[b * uniq:(b + 1) * uniq]])
        vv = torch.cat([vs, vc.view(-1, Hkv, D)[b * uniq:(b + 1) * uniq]])
        logits = torch.einsum("hd,khd->hk", q[b].float(),
                              kv.float().repeat_interleave(G, 1)) / math.sqrt(D)
        ref = torch.einsum("hk,khd->hd", torch.softmax(logits, -1),
                           vv.float().repeat_interleave(G, 1))
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.gpu
def test_append_paged_mla_kv_cache():
    torch.manual_seed(1)
    B, page = 2, 16
    pages_per = 4
    npages = B * pages_per
    ckv_cache = torch.zeros(npages, page, 512, dtype=torch.bfloat16,
                            device="cuda")
    kpe_cache = torch.zeros(npages, page, 64, dtype=torch.bfloat16,
                            device="cuda")
    kv_indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                             device="cuda")
    kv_indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lens = torch.tensor([20, 35], dtype=torch.int32, device="cuda")
    append_indptr = torch.tensor([0, 3, 5], dtype=torch.int32, device="cuda")
    bi, pos = fi.get_batch_indices_positions(append_indptr, lens, 5)
    ckv = torch.randn(5, 512, dtype=torch.bfloat16, device="cuda")
    kpe = torch.randn(5, 64, dtype=torch.bfloat16, device="cuda")
    last = ((lens - 1) % page + 1).int()
    fi.append_paged_mla_kv_cache(ckv, kpe, bi, pos, ckv_cache, kpe_cache,
                                 kv_indices, kv_indptr, last)
    # request 0 appended 3 rows at positions 17,18,19
    for j, p in enumerate([17, 18, 19]):
        pg = int(kv_indices[int(kv_indptr[0]) + p // page])
        torch.testing.assert_close(ckv_cache[pg, p % page], ckv[j])
        torch.testing.assert_close(kpe_cache[pg, p % page], kpe[j])


@pytest.mark.gpu
def test_recurrent_kda_batch_form():
    torch.manual_seed(2)
    B, L, H, D = 2, 10, 2, 64
    q = torch.randn(B, L, H, D, device="cuda").bfloat16()
    k = torch.nn.functional.normalize(
        torch.randn(B, L, H, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(B, L, H, D, device="cuda") / 4).bfloat16()
    g = torch.rand(B, L, H, D, device="cuda") * 0.8 + 0.1
    beta = torch.rand(B, L, H, device="cuda")
    o, S = fi.recurrent_kda(q, k, v, g, beta, output_final_state=True)
    assert o.shape == (B, L, H, D) and S.shape == (B, H, D, D)
    assert o.float().isfinite().all()


@pytest.mark.gpu
def test_dit_fused_layernorms():
    torch.manual_seed(3)
    B, S, H = 2, 8, 256
    x = torch.randn(B, S, H, dtype=torch.bfloat16, device="cuda")
    res = torch.randn_like(x)
    gate = torch.randn(B, 1, H, device="cuda")
    gamma = torch.randn(H, device="cuda")
    beta = torch.randn(H, device="cuda")
    r, n = fi.fused_dit_gate_residual_layernorm_gamma_beta(
        x, res, gate, gamma, beta)
    ref_r = res.float() + x.float() * gate
    ref_n = torch.nn.functional.layer_norm(ref_r, [H], gamma, beta, 1e-6)
    torch.testing.assert_close(r.float(), ref_r, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(n.float(), ref_n, atol=2e-2, rtol=2e-2)
    scale = torch.randn(B, 1, H, device="cuda")
    shift = torch.randn(B, 1, H, device="cuda")
    r2, n2 = fi.fused_dit_residual_layernorm_scale_shift(x, scale, shift,
                                                         residual=res)
    ref_r2 = res.float() + x.float()
    ref_n2 = torch.nn.functional.layer_norm(ref_r2, [H], None, None, 1e-6) \
        * (1 + scale) + shift
    torch.testing.assert_close(n2.float(), ref_n2, atol=2e-2, rtol=2e-2)


@pytest.mark.gpu
def test_trtllm_bf16_moe_routed():
    torch.manual_seed(4)
    T, Hd, I, E, K = 64, 256, 512, 4, 2
    x = torch.randn(T, Hd, dtype=torch.bfloat16, device="cuda")
    w13 = torch.randn(E, 2 * I, Hd, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, Hd, I, dtype=torch.bfloat16, device="cuda") / 16
    logits = torch.randn(T, E, device="cuda")
    out = fi.trtllm_bf16_moe(logits, x, w13, w2, K)
    assert out.shape == (T, Hd) and out.float().isfinite().all()
    # pre-routed variant agrees with the same routing
    from flashinfer_amd.fused_moe import _route, RoutingMethodType
    w, ids = _route(logits, K, RoutingMethodType.Renormalize)
    out2 = fi.trtllm_bf16_routed_moe(x, ids, w, w13, w2)
    torch.testing.assert_close(out.float(), out2.float())
