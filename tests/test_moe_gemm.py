"""GPU tests: grouped GEMM + fused MoE vs torch references."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_group_gemm_matches_loop():
    from flashinfer_amd._lib import get_ext

    torch.manual_seed(0)
    E, N, K = 4, 512, 256
    m_sizes = [0, 200, 128, 37]
    M = sum(m_sizes)
    m_indptr = torch.zeros(E + 1, dtype=torch.int32, device="cuda")
    m_indptr[1:] = torch.cumsum(torch.tensor(m_sizes, device="cuda"), 0).int()
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") / 8
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") / 8
    c = torch.zeros(M, N, dtype=torch.bfloat16, device="cuda")
    get_ext().group_gemm_nt(a, w, c, m_indptr, None, (M + 127) // 128 + 1)
    for e in range(E):
        s, t = int(m_indptr[e]), int(m_indptr[e + 1])
        if s == t:
            continue
        ref = a[s:t].float() @ w[e].float().t()
        torch.testing.assert_close(c[s:t].float(), ref, atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("T,E,k", [(64, 8, 2), (333, 16, 4)])
def test_fused_moe_matches_dense(T, E, k):
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax

    torch.manual_seed(0)
    H, inter = 256, 512
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 16
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    out = fused_moe(x, w13, w2, weights, ids)

    ref = torch.zeros(T, H, dtype=torch.float32, device="cuda")
    for t in range(T):
        for j in range(k):
            e = int(ids[t, j])
            h = x[t].float() @ w13[e].float().t()
            g, u = h.chunk(2)
            act = torch.nn.functional.silu(g) * u
            ref[t] += weights[t, j].float() * (act @ w2[e].float().t())
    torch.testing.assert_close(out.float(), ref, atol=8e-2, rtol=8e-2)


def test_dsv3_routing_properties():
    from flashinfer_amd.fused_moe import dsv3_routing

    torch.manual_seed(0)
    T, E, ng, tkg, k = 32, 256, 8, 4, 8
    logits = torch.randn(T, E, device="cuda")
    w, ids = dsv3_routing(logits, k, ng, tkg, routed_scaling_factor=2.5)
    assert ids.shape == (T, k) and w.shape == (T, k)
    # selected experts come from at most tkg groups
    groups = (ids // (E // ng)).long()
    for t in range(T):
        assert groups[t].unique().numel() <= tkg
    torch.testing.assert_close(w.sum(-1), torch.full((T,), 2.5, device="cuda"),
                               atol=1e-4, rtol=1e-4)


def test_fused_moe_fp8_matches_dequant():
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import per_block_quant_fp8

    torch.manual_seed(0)
    T, E, k, H, inter = 128, 4, 2, 256, 256
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 8
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 8
    w13_q, w13_s = per_block_quant_fp8(w13)
    w2_q, w2_s = per_block_quant_fp8(w2)
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    out_fp8 = fused_moe(x, w13_q, w2_q, weights, ids,
                        w13_scale=w13_s, w2_scale=w2_s)
    out_bf16 = fused_moe(x, w13, w2, weights, ids)
    # fp8 ~2 decimal digits; compare against the bf16 path loosely
    err = (out_fp8.float() - out_bf16.float()).abs().mean() / out_bf16.float().abs().mean().clamp(min=1e-6)
    assert err < 0.12, f"relative error {err}"  # two fp8 quant stages vs bf16


def test_segment_gemm_wrapper():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    ws = torch.empty(8 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    seg = fi.SegmentGEMMWrapper(ws)
    K, N = 128, 256
    seg_lens = torch.tensor([17, 0, 40])
    M = int(seg_lens.sum())
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") / 4
    weights = torch.randn(5, N, K, dtype=torch.bfloat16, device="cuda") / 4
    w_idx = torch.tensor([4, 0, 2], dtype=torch.int32, device="cuda")
    y = seg.run(x, weights, 3, seg_lens=seg_lens, weight_indices=w_idx)
    off = 0
    for i, L in enumerate(seg_lens.tolist()):
        if L:
            ref = x[off:off + L].float() @ weights[int(w_idx[i])].float().t()
            torch.testing.assert_close(y[off:off + L].float(), ref,
                                       atol=5e-2, rtol=5e-2)
        off += L


@pytest.mark.gpu
def test_grouped_mm_bf16():
    from flashinfer_amd.grouped_mm import grouped_mm_bf16

    torch.manual_seed(0)
    E, K, N = 3, 256, 128
    ms = [128, 0, 200]
    m_indptr = torch.tensor([0, 128, 128, 328], dtype=torch.int32, device="cuda")
    a = torch.randn(328, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") / 8
    out = grouped_mm_bf16(a, b, m_indptr)
    for g in range(E):
        s, e = int(m_indptr[g]), int(m_indptr[g + 1])
        if s == e:
            continue
        ref = a[s:e].float() @ b[g].float().t()
        torch.testing.assert_close(out[s:e].float(), ref, atol=2e-1, rtol=2e-2)


@pytest.mark.gpu
def test_deep_gemm_contiguous_and_masked():
    import flashinfer_amd as fi
    from flashinfer_amd.deep_gemm import (
        m_grouped_fp8_gemm_nt_contiguous, m_grouped_fp8_gemm_nt_masked)

    torch.manual_seed(1)
    E, K, N = 2, 256, 128
    b, sfb = fi.per_block_quant_fp8(torch.randn(E, N, K, device="cuda") / 4)
    # contiguous: rows 0..127 expert 0, 128..255 expert 1
    M = 256
    af = torch.randn(M, K, device="cuda") / 4
    a, sfa = fi.per_token_group_quant_fp8(af.bfloat16())  # [M, K/128]
    m_indices = torch.repeat_interleave(
        torch.arange(E, device="cuda"), M // E).int()
    d = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    m_grouped_fp8_gemm_nt_contiguous((a, sfa), (b, sfb), d, m_indices)
    for g in range(E):
        s, e = g * (M // E), (g + 1) * (M // E)
        bsc = sfb[g].repeat_interleave(128, 0).repeat_interleave(128, 1)  # [K, N]
        b_deq = b[g].float() * bsc.t()[:, :K]
        ref = (a[s:e].float() * sfa[s:e].repeat_interleave(128, 1)) @ \
              b_deq.t()
        torch.testing.assert_close(d[s:e].float(), ref, atol=2e-1, rtol=5e-2)

    # masked
    m_max = 64
    a2f = torch.randn(E, m_max, K, device="cuda") / 4
    a2, sfa2_flat = fi.per_token_group_quant_fp8(
        a2f.reshape(E * m_max, K).bfloat16())
    a2 = a2.view(E, m_max, K)
    sfa2 = sfa2_flat.view(E, m_max, -1)
    masked_m = torch.tensor([40, 10], dtype=torch.int32, device="cuda")
    d2 = torch.zeros(E, m_max, N, dtype=torch.bfloat16, device="cuda")
    m_grouped_fp8_gemm_nt_masked((a2, sfa2), (b, sfb), d2, masked_m, 40)
    for g in range(E):
        mv = int(masked_m[g])
        bsc = sfb[g].repeat_interleave(128, 0).repeat_interleave(128, 1)
        b_deq = b[g].float() * bsc.t()[:, :K]
        ref = (a2[g, :mv].float() *
               sfa2[g, :mv].repeat_interleave(128, 1)) @ b_deq.t()
        torch.testing.assert_close(d2[g, :mv].float(), ref, atol=2e-1, rtol=5e-2)


@pytest.mark.gpu
def test_moe_routing_kernels_vs_torch():
    """Fused routing kernels vs the torch reference formulas."""
    import importlib
    fm = importlib.import_module("flashinfer_amd.fused_moe")

    torch.manual_seed(0)
    for T, E, k in [(64, 8, 2), (128, 64, 8), (33, 256, 8)]:
        logits = torch.randn(T, E, device="cuda")
        w_k, i_k = fm.moe_topk_softmax(logits, k)
        # torch reference
        probs = torch.softmax(logits.float(), dim=-1)
        w_r, i_r = torch.topk(probs, k, dim=-1)
        w_r = w_r / w_r.sum(-1, keepdim=True).clamp(min=1e-20)
        assert torch.equal(torch.sort(i_k, -1).values,
                           torch.sort(i_r.int(), -1).values)
        torch.testing.assert_close(w_k.sort(-1).values, w_r.sort(-1).values,
                                   atol=1e-5, rtol=1e-5)


@pytest.mark.gpu
def test_dsv3_routing_kernel_vs_torch():
    import importlib
    fm = importlib.import_module("flashinfer_amd.fused_moe")

    torch.manual_seed(1)
    T, E, k, n_group, topk_group = 77, 256, 8, 8, 4
    logits = torch.randn(T, E, device="cuda")
    bias = torch.randn(E, device="cuda") * 0.1
    w_k, i_k = fm.dsv3_routing(logits, k, n_group, topk_group, 2.5, bias)
    # force the torch path by moving to CPU
    w_r, i_r = fm.dsv3_routing(logits.cpu(), k, n_group, topk_group, 2.5,
                               bias.cpu())
    assert torch.equal(torch.sort(i_k.cpu(), -1).values,
                       torch.sort(i_r.int(), -1).values)
    torch.testing.assert_close(w_k.cpu().sort(-1).values,
                               w_r.sort(-1).values, atol=1e-5, rtol=1e-5)


@pytest.mark.gpu
def test_moe_build_permute_kernel():
    from flashinfer_amd.fused_moe import _build_permute

    torch.manual_seed(2)
    T, k, E = 512, 2, 8
    ids = torch.randint(0, E, (T, k), dtype=torch.int32, device="cuda")
    m_indptr, token_of_copy, inv = _build_permute(ids, E)
    flat = ids.reshape(-1).long()
    counts = torch.bincount(flat, minlength=E)
    ref_indptr = torch.zeros(E + 1, dtype=torch.int64, device="cuda")
    ref_indptr[1:] = counts.cumsum(0)
    assert torch.equal(m_indptr.long(), ref_indptr)
    # every permuted row's expert matches its segment; inv is a bijection
    e_of_pos = flat[torch.argsort(inv.long())]  # expert of copy at pos p
    for e in range(E):
        seg = e_of_pos[int(ref_indptr[e]):int(ref_indptr[e + 1])]
        assert (seg == e).all()
    assert torch.equal(inv.long().sort().values,
                       torch.arange(T * k, device="cuda"))
    # token_of_copy consistency
    tok = (torch.arange(T * k, device="cuda") // k)
    assert torch.equal(token_of_copy.long()[inv.long()], tok)


@pytest.mark.gpu
def test_silu_mul_quant_and_gather_quant():
    from flashinfer_amd._lib import get_ext

    torch.manual_seed(3)
    R, I = 64, 256
    h = torch.randn(R, 2 * I, dtype=torch.bfloat16, device="cuda")
    q = torch.empty(R, I, dtype=torch.uint8, device="cuda")
    s = torch.empty(I // 128, R, dtype=torch.float32, device="cuda")
    get_ext().silu_mul_quant_run(h, q, s, False)
    gate, up = h.float().chunk(2, dim=-1)
    ref = torch.nn.functional.silu(gate) * up
    deq = q.view(torch.float8_e4m3fn).float() * \
        s.t().repeat_interleave(128, dim=1)
    torch.testing.assert_close(deq, ref, atol=0.08, rtol=0.08)

    x = torch.randn(16, 256, dtype=torch.bfloat16, device="cuda")
    toc = torch.randint(0, 16, (32,), dtype=torch.int32, device="cuda")
    gq = torch.empty(32, 256, dtype=torch.uint8, device="cuda")
    gs = torch.empty(2, 32, dtype=torch.float32, device="cuda")
    get_ext().gather_quant_run(x, toc, gq, gs)
    ref2 = x.float()[toc.long()]
    deq2 = gq.view(torch.float8_e4m3fn).float() * \
        gs.t().repeat_interleave(128, dim=1)
    torch.testing.assert_close(deq2, ref2, atol=0.08, rtol=0.08)


@pytest.mark.gpu
def test_moe_e2e_mxfp8():
    """MX-fp8 MoE (hardware e8m0 scales) vs a torch reference."""
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import per_block_quant_mxfp8

    torch.manual_seed(4)
    T, H, inter, E, k = 128, 512, 512, 8, 2
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 8
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 8
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    w13_q, w13_s = per_block_quant_mxfp8(w13)
    w2_q, w2_s = per_block_quant_mxfp8(w2)
    assert w13_s.dtype == torch.uint8
    out = fused_moe(x, w13_q, w2_q, weights, ids,
                    w13_scale=w13_s, w2_scale=w2_s)
    ref = torch.zeros(T, H, device="cuda")
    xf = x.float()
    for t in range(T):
        for j in range(k):
            e = int(ids[t, j])
            h1 = xf[t] @ w13[e].float().t()
            act = torch.nn.functional.silu(h1[:inter]) * h1[inter:]
            ref[t] += float(weights[t, j]) * (act @ w2[e].float().t())
    # pow2 scales cost up to ~1 bit of quant precision vs f32 groupwise
    torch.testing.assert_close(out.float(), ref, atol=0.5, rtol=0.2)


@pytest.mark.gpu
def test_dsv3_router_gemms():
    """DSv3 skinny router GEMMs (dsv3_ops re-exports) vs torch matmul."""
    from flashinfer_amd import dsv3_ops

    torch.manual_seed(0)
    for M in (1, 16):
        a = torch.randn(M, 7168, dtype=torch.bfloat16, device="cuda")
        for N, fn in ((128, dsv3_ops.mm_M1_16_K7168_N128),
                      (256, dsv3_ops.mm_M1_16_K7168_N256)):
            b = torch.randn(7168, N, dtype=torch.bfloat16, device="cuda")
            out = fn(a, b)
            ref = (a.float() @ b.float())
            torch.testing.assert_close(out.float(), ref, atol=2e-1, rtol=2e-2)
