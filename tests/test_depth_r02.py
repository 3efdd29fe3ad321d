"""Round-2 depth tests: large-shape ragged prefill sweeps,
fp8-KV long-context accuracy, MoE e2e at 64 experts, hipGraph
capture-replan-replay, GPU sampling determinism."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ref_attn(q, k, v, causal):
    M, Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    qf = q.float().transpose(0, 1)
    kf = k.float().repeat_interleave(g, 1).transpose(0, 1)
    vf = v.float().repeat_interleave(g, 1).transpose(0, 1)
    logits = qf @ kf.transpose(-1, -2) / math.sqrt(D)
    if causal:
        qpos = torch.arange(M, device=q.device)[:, None]
        kpos = torch.arange(L, device=q.device)[None, :]
        logits = logits.masked_fill((kpos > qpos + L - M)[None], float("-inf"))
    return (torch.softmax(logits, -1) @ vf).transpose(0, 1)


@pytest.mark.parametrize("qo,kv,Hq,Hkv,causal", [
    (4096, 4096, 8, 1, True),        # long single-request ragged
    (2048, 2048, 16, 2, False),
    (1, 32768, 8, 8, False),         # extreme split-KV
    (8192, 8192, 4, 4, True),
])
def test_ragged_prefill_large_shapes(qo, kv, Hq, Hkv, causal):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    D = 128
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda") / 2
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda") / 2
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda") / 2
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, kv], dtype=torch.int32, device="cuda")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=causal,
           q_data_type=torch.bfloat16)
    out = w.run(q, k, v)
    # spot-check a contiguous row slice (full fp32 reference at 8k x 32k
    # would dominate suite time)
    rows = slice(0, min(qo, 256))
    ref = _ref_attn(q[rows], k, v, causal) if qo <= 256 else None
    if ref is None:
        # reference over the selected rows only (causal offset preserved)
        qf = q.float()[rows].transpose(0, 1)
        kf = k.float().repeat_interleave(Hq // Hkv, 1).transpose(0, 1)
        vf = v.float().repeat_interleave(Hq // Hkv, 1).transpose(0, 1)
        logits = qf @ kf.transpose(-1, -2) / math.sqrt(D)
        if causal:
            qpos = torch.arange(256, device="cuda")[:, None]
            kpos = torch.arange(kv, device="cuda")[None, :]
            logits = logits.masked_fill((kpos > qpos + kv - qo)[None],
                                        float("-inf"))
        ref = (torch.softmax(logits, -1) @ vf).transpose(0, 1)
    torch.testing.assert_close(out[rows].float(), ref, atol=3e-2, rtol=3e-2)


def test_fp8_kv_prefill_long_context():
    """fp8 (e4m3) KV cache accuracy at kv=8192 vs the dequantized fp32
    reference (reference prefill.cuh:1150 repack path)."""
    import flashinfer_amd as fi

    torch.manual_seed(1)
    qo, kv, Hq, Hkv, D, page = 128, 8192, 8, 2, 128, 16
    pages = kv // page
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda") / 2
    kc = (torch.randn(pages, page, Hkv, D, device="cuda") / 2).to(
        torch.float8_e4m3fn)
    vc = (torch.randn(pages, page, Hkv, D, device="cuda") / 2).to(
        torch.float8_e4m3fn)
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, pages], dtype=torch.int32, device="cuda")
    kv_indices = torch.arange(pages, dtype=torch.int32, device="cuda")
    lpl = torch.tensor([page], dtype=torch.int32, device="cuda")
    w.plan(qo_indptr, kv_indptr, kv_indices, lpl, Hq, Hkv, D, page,
           causal=False, q_data_type=torch.bfloat16,
           kv_data_type=torch.float8_e4m3fn)
    out = w.run(q, (kc, vc))
    kd = kc.view(torch.float8_e4m3fn).float().view(kv, Hkv, D).to(torch.bfloat16)
    vd = vc.view(torch.float8_e4m3fn).float().view(kv, Hkv, D).to(torch.bfloat16)
    ref = _ref_attn(q, kd, vd, False)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_moe_e2e_64_experts():
    """fp8 fused MoE vs a per-expert torch reference at 64 experts
    (exercises the flat-tile grouped GEMM path, E >= 16)."""
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import per_block_quant_fp8

    torch.manual_seed(2)
    T, H, inter, E, k = 256, 512, 512, 64, 4
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 8
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 8
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    w13_q, w13_s = per_block_quant_fp8(w13)
    w2_q, w2_s = per_block_quant_fp8(w2)
    out = fused_moe(x, w13_q, w2_q, weights, ids,
                    w13_scale=w13_s, w2_scale=w2_s)
    # torch reference in fp32 with the true (unquantized) weights
    ref = torch.zeros(T, H, device="cuda")
    xf = x.float()
    for t in range(T):
        for j in range(k):
            e = int(ids[t, j])
            h1 = xf[t] @ w13[e].float().t()
            act = torch.nn.functional.silu(h1[:inter]) * h1[inter:]
            ref[t] += float(weights[t, j]) * (act @ w2[e].float().t())
    torch.testing.assert_close(out.float(), ref, atol=0.35, rtol=0.12)


def test_decode_graph_replan_replay():
    """hipGraph capture -> REPLAN with new kv lengths -> replay must reflect
    the new plan (reference CUDAGraph wrapper contract, decode.py:1478)."""
    import flashinfer_amd as fi

    torch.manual_seed(3)
    Hq, Hkv, D, page, bs = 32, 8, 128, 16, 4
    max_pages = bs * 64
    indptr_buf = torch.zeros(bs + 1, dtype=torch.int32, device="cuda")
    indices_buf = torch.zeros(max_pages, dtype=torch.int32, device="cuda")
    lpl_buf = torch.zeros(bs, dtype=torch.int32, device="cuda")
    kc = torch.randn(max_pages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(max_pages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.CUDAGraphBatchDecodeWithPagedKVCacheWrapper(
        ws, indptr_buf, indices_buf, lpl_buf, "NHD")

    def plan_lens(kv_lens):
        pages_per = [(L + page - 1) // page for L in kv_lens]
        ip = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                          dtype=torch.int32)
        idx = torch.arange(int(ip[-1]), dtype=torch.int32)
        lp = torch.tensor([(L - 1) % page + 1 for L in kv_lens],
                          dtype=torch.int32)
        w.plan(ip, idx, lp, Hq, Hkv, D, page, q_data_type=torch.bfloat16)

    def ref_for(kv_lens):
        res = []
        pos = 0
        for b, L in enumerate(kv_lens):
            npg = (L + page - 1) // page
            kk = kc[pos:pos + npg].view(-1, Hkv, D)[:L]
            vv = vc[pos:pos + npg].view(-1, Hkv, D)[:L]
            g = Hq // Hkv
            lg = torch.einsum(
                "hd,lhd->hl", q[b].float(),
                kk.float().repeat_interleave(g, 1)) / math.sqrt(D)
            res.append(torch.einsum(
                "hl,lhd->hd", torch.softmax(lg, -1),
                vv.float().repeat_interleave(g, 1)))
            pos += npg
        return torch.stack(res)

    lens1 = [100, 200, 64, 333]
    plan_lens(lens1)
    w.run(q, (kc, vc), out=out)  # warm
    torch.cuda.synchronize()
    g_ = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g_):
        w.run(q, (kc, vc), out=out)
    g_.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref_for(lens1), atol=3e-2,
                               rtol=3e-2)
    # replan with DIFFERENT lengths into the same fixed buffers, replay
    lens2 = [555, 64, 900, 16]
    plan_lens(lens2)
    torch.cuda.synchronize()
    g_.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref_for(lens2), atol=3e-2,
                               rtol=3e-2)


def test_sampling_gpu_determinism():
    """Bitwise run-to-run determinism of the GPU sampling suite (reference
    deterministic scans, sampling.cuh:195)."""
    import flashinfer_amd as fi

    torch.manual_seed(4)
    probs = torch.softmax(torch.randn(64, 32000, device="cuda"), -1)
    gen1 = torch.Generator("cuda").manual_seed(42)
    ids1 = fi.top_k_sampling_from_probs(probs, 50, generator=gen1)
    gen2 = torch.Generator("cuda").manual_seed(42)
    ids2 = fi.top_k_sampling_from_probs(probs, 50, generator=gen2)
    assert torch.equal(ids1, ids2)
    r1 = fi.top_p_renorm_probs(probs, 0.9)
    r2 = fi.top_p_renorm_probs(probs, 0.9)
    assert torch.equal(r1, r2)
    g3 = torch.Generator("cuda").manual_seed(7)
    s1 = fi.sampling_from_probs(probs, generator=g3)
    g4 = torch.Generator("cuda").manual_seed(7)
    s2 = fi.sampling_from_probs(probs, generator=g4)
    assert torch.equal(s1, s2)


def test_autotune_decode_route():
    """Under autotune(), the decode wrapper profiles its three kernel shapes
    on the real inputs and caches a winner for the bucketed shape key."""
    import flashinfer_amd as fi
    from flashinfer_amd import autotuner

    torch.manual_seed(5)
    Hq, Hkv, D, page, bs, kv = 64, 8, 128, 16, 16, 1024
    pages_per = kv // page
    indptr = torch.arange(0, (bs + 1) * pages_per, pages_per,
                          dtype=torch.int32, device="cuda")
    indices = torch.randperm(bs * pages_per, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = torch.randn(bs * pages_per, page, Hkv, D, dtype=torch.bfloat16,
                     device="cuda")
    vc = torch.randn(bs * pages_per, page, Hkv, D, dtype=torch.bfloat16,
                     device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    autotuner._cache.clear()
    try:
        _run_autotune_body(fi, autotuner, ws, indptr, indices, lpl, Hq, Hkv,
                           D, page, q, kc, vc)
    finally:
        # the route cache is global state: scrub it so later tests'
        # routing assertions see the heuristic, not this test's winner
        for k in [k for k in autotuner._cache if k.startswith("decode_route")]:
            del autotuner._cache[k]


def _run_autotune_body(fi, autotuner, ws, indptr, indices, lpl, Hq, Hkv, D,
                       page, q, kc, vc):
    with autotuner.autotune():
        w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
        w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
               q_data_type=torch.bfloat16)
        out = w.run(q, (kc, vc))
    keys = [k for k in autotuner._cache if k.startswith("decode_route")]
    assert keys, "route tactic must be cached"
    # cached winner is used on the next plan (no profiling outside ctx)
    w2 = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w2.plan(indptr, indices, lpl, Hq, Hkv, D, page,
            q_data_type=torch.bfloat16)
    out2 = w2.run(q, (kc, vc))
    # numerics unchanged regardless of the winning route
    torch.testing.assert_close(out.float(), out2.float(), atol=2e-2, rtol=2e-2)
