"""GPU numerics: single/batch decode attention vs PyTorch SDPA fp32 reference."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def sdpa_ref(q, k, v, sm_scale=None, soft_cap=0.0, window_left=-1):
    # q [Hq, D]; k/v [L, Hkv, D] -> out [Hq, D] fp32
    Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    qf = q.float()
    kf = k.float().repeat_interleave(g, dim=1)  # [L, Hq, D]
    vf = v.float().repeat_interleave(g, dim=1)
    scale = sm_scale if sm_scale is not None else 1 / math.sqrt(D)
    logits = torch.einsum("hd,lhd->hl", qf, kf) * scale
    if soft_cap > 0:
        logits = soft_cap * torch.tanh(logits / soft_cap)
    if window_left >= 0:
        mask = torch.arange(L, device=q.device) >= (L - 1 - window_left)
        logits = logits.masked_fill(~mask[None, :], float("-inf"))
    p = torch.softmax(logits, dim=-1)
    return torch.einsum("hl,lhd->hd", p, vf)


@pytest.mark.parametrize("kv_len", [1, 54, 2048, 8190])
@pytest.mark.parametrize("Hq,Hkv", [(32, 8), (8, 8), (8, 1)])
@pytest.mark.parametrize("D", [128, 64])
def test_single_decode(kv_len, Hq, Hkv, D):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    q = torch.randn(Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    out = fi.single_decode_with_kv_cache(q, k, v)
    ref = sdpa_ref(q, k, v)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_single_decode_soft_cap_window():
    import flashinfer_amd as fi

    torch.manual_seed(1)
    q = torch.randn(8, 128, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(1024, 2, 128, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(1024, 2, 128, dtype=torch.bfloat16, device="cuda")
    out = fi.single_decode_with_kv_cache(q, k, v, logits_soft_cap=30.0, window_left=127)
    ref = sdpa_ref(q, k, v, soft_cap=30.0, window_left=127)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("page_size", [1, 16])
@pytest.mark.parametrize("kv_layout", ["NHD", "HND"])
# (64, 8, 128) exercises the MFMA split route with the same-XCD in-kernel
# merge (batch 5 x 8 kv_heads % 8 == 0) including its LSE epilogue
@pytest.mark.parametrize("Hq,Hkv,D", [(32, 8, 128), (4, 4, 64), (64, 8, 128)])
def test_batch_decode(page_size, kv_layout, Hq, Hkv, D):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    kv_lens = [1, 17, 500, 2049, 128]
    batch = len(kv_lens)
    pages_per = [(L + page_size - 1) // page_size for L in kv_lens]
    indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                          dtype=torch.int32, device="cuda")
    npages = int(indptr[-1])
    indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last_page_len = torch.tensor(
        [(L - 1) % page_size + 1 for L in kv_lens], dtype=torch.int32, device="cuda"
    )
    if kv_layout == "NHD":
        shape = (npages, page_size, Hkv, D)
    else:
        shape = (npages, Hkv, page_size, D)
    k_cache = torch.randn(shape, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(shape, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(batch, Hq, D, dtype=torch.bfloat16, device="cuda")

    ws = torch.empty(64 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    wrapper = fi.BatchDecodeWithPagedKVCacheWrapper(ws, kv_layout)
    wrapper.plan(indptr, indices, last_page_len, Hq, Hkv, D, page_size,
                 q_data_type=torch.bfloat16)
    out, lse = wrapper.run(q, (k_cache, v_cache), return_lse=True)

    # reference: gather pages per request
    for b in range(batch):
        L = kv_lens[b]
        toks = []
        for p in range(pages_per[b]):
            page = int(indices[int(indptr[b]) + p])
            n = min(page_size, L - p * page_size)
            if kv_layout == "NHD":
                toks.append((k_cache[page, :n], v_cache[page, :n]))
            else:
                toks.append(
                    (k_cache[page, :, :n].transpose(0, 1),
                     v_cache[page, :, :n].transpose(0, 1))
                )
        kk = torch.cat([t[0] for t in toks], 0)
        vv = torch.cat([t[1] for t in toks], 0)
        ref = sdpa_ref(q[b], kk, vv)
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)
        # lse check (base-2)
        g = Hq // Hkv
        kf = kk.float().repeat_interleave(g, dim=1)
        logits = torch.einsum("hd,lhd->hl", q[b].float(), kf) / math.sqrt(D)
        ref_lse = torch.logsumexp(logits, -1) / math.log(2)
        torch.testing.assert_close(lse[b], ref_lse, atol=2e-2, rtol=2e-2)


def test_decode_with_empty_request():
    """A request with zero KV pages must produce zero output, not NaN."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    # req0: 2 pages, req1: EMPTY, req2: 1 page
    indptr = torch.tensor([0, 2, 2, 3], dtype=torch.int32, device="cuda")
    indices = torch.arange(3, dtype=torch.int32, device="cuda")
    lpl = torch.tensor([page, 0, 5], dtype=torch.int32, device="cuda")
    kc = torch.randn(3, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(3, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(3, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(16 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out, lse = w.run(q, (kc, vc), return_lse=True)
    assert out[1].abs().max() == 0           # empty request -> zeros
    assert torch.isinf(lse[1]).all() and (lse[1] < 0).all()
    assert out[0].isfinite().all() and out[2].isfinite().all()


@pytest.mark.parametrize("Hq,Hkv", [(8, 8), (16, 2)])
def test_decode_head_dim_256(Hq, Hkv):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    D, page, bs, kv = 256, 16, 4, 333
    pages_per = (kv + page - 1) // page
    npages = bs * pages_per
    indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                          device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32,
                     device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = w.run(q, (kc, vc))
    import math
    G = Hq // Hkv
    for b in range(bs):
        kb = kc.view(-1, Hkv, D)[b * pages_per * page:][:kv].float()
        vb = vc.view(-1, Hkv, D)[b * pages_per * page:][:kv].float()
        logits = torch.einsum("hd,khd->hk", q[b].float(),
                              kb.repeat_interleave(G, 1)) / math.sqrt(D)
        ref = torch.einsum("hk,khd->hd", torch.softmax(logits, -1),
                           vb.repeat_interleave(G, 1))
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)


# ---------------- fused whole-request decode (short-kv shape) ----------------

def _paged(bs, kv_lens, Hkv, D, page, dtype=torch.bfloat16):
    pages_per = [(L + page - 1) // page for L in kv_lens]
    indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                          dtype=torch.int32, device="cuda")
    npages = int(indptr[-1])
    indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    lpl = torch.tensor([(L - 1) % page + 1 if L else 0 for L in kv_lens],
                       dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    return indptr, indices, lpl, kc, vc


def _gather(indptr, indices, kv_lens, kc, vc, b, page):
    L = kv_lens[b]
    toks_k, toks_v = [], []
    for p in range((L + page - 1) // page):
        pg = int(indices[int(indptr[b]) + p])
        n = min(page, L - p * page)
        toks_k.append(kc[pg, :n])
        toks_v.append(vc[pg, :n])
    return torch.cat(toks_k, 0), torch.cat(toks_v, 0)


@pytest.mark.parametrize("Hq,Hkv,D", [(64, 8, 128), (32, 8, 128), (8, 8, 64),
                                      (8, 2, 128), (16, 4, 256)])
def test_decode_fused_routing_and_numerics(Hq, Hkv, D):
    """Short-kv plans must route to the fused whole-request kernel and match
    the fp32 reference (covers the BASELINE bs=16/kv=1024 GQA-8 config)."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    page = 16
    kv_lens = [1024, 1, 777, 1024, 16, 300] + [1024] * 10
    bs = len(kv_lens)
    indptr, indices, lpl, kc, vc = _paged(bs, kv_lens, Hkv, D, page)
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w._fused, "short-kv plan must pick the fused kernel"
    out, lse = w.run(q, (kc, vc), return_lse=True)
    for b in [0, 1, 2, 4, 5]:
        kk, vv = _gather(indptr, indices, kv_lens, kc, vc, b, page)
        ref = sdpa_ref(q[b], kk, vv)
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)
        g = Hq // Hkv
        kf = kk.float().repeat_interleave(g, dim=1)
        logits = torch.einsum("hd,lhd->hl", q[b].float(), kf) / math.sqrt(D)
        ref_lse = torch.logsumexp(logits, -1) / math.log(2)
        torch.testing.assert_close(lse[b], ref_lse, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("soft_cap,window", [(30.0, -1), (0.0, 127), (30.0, 127)])
def test_decode_fused_softcap_window(soft_cap, window):
    import flashinfer_amd as fi

    torch.manual_seed(1)
    Hq, Hkv, D, page = 32, 8, 128, 16
    kv_lens = [513, 1024, 64]
    bs = len(kv_lens)
    indptr, indices, lpl, kc, vc = _paged(bs, kv_lens, Hkv, D, page)
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           logits_soft_cap=soft_cap or None, window_left=window,
           q_data_type=torch.bfloat16)
    assert w._fused
    out = w.run(q, (kc, vc))
    for b in range(bs):
        kk, vv = _gather(indptr, indices, kv_lens, kc, vc, b, page)
        ref = sdpa_ref(q[b], kk, vv, soft_cap=soft_cap, window_left=window)
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)


def test_decode_fused_batch_invariant():
    """Batch-invariance contract (reference batch-invariant FA2 mode):
    with disable_split_kv the per-request computation depends only on the
    request's own kv_len, so a request's output is bitwise identical
    regardless of the other requests in the batch. (The AUTO route sizes
    its cross-WG split to fill the chip — batch-dependent by design — so
    the bitwise guarantee is tied to the deterministic plan options, as in
    the reference; auto is still run-to-run deterministic.)"""
    import flashinfer_amd as fi

    torch.manual_seed(2)
    Hq, Hkv, D, page = 32, 8, 128, 16
    kv_lens = [1024, 300, 2000]
    indptr, indices, lpl, kc, vc = _paged(3, kv_lens, Hkv, D, page)
    q = torch.randn(3, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16,
           disable_split_kv=True)
    assert w._fused
    full = w.run(q, (kc, vc))
    # rerun -> bitwise identical
    again = w.run(q, (kc, vc))
    assert torch.equal(full, again)
    # request 0 alone -> bitwise identical to its row in the full batch
    w2 = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w2.plan(indptr[:2], indices, lpl[:1], Hq, Hkv, D, page,
            q_data_type=torch.bfloat16, disable_split_kv=True)
    assert w2._fused
    solo = w2.run(q[:1], (kc, vc))
    assert torch.equal(solo[0], full[0])
    # AUTO route: run-to-run deterministic and numerically equal across
    # batch compositions (split differs -> not bitwise)
    wa = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    wa.plan(indptr, indices, lpl, Hq, Hkv, D, page,
            q_data_type=torch.bfloat16)
    fa = wa.run(q, (kc, vc))
    assert torch.equal(fa, wa.run(q, (kc, vc)))
    torch.testing.assert_close(fa[0].float(), full[0].float(),
                               atol=3e-3, rtol=3e-3)


def test_decode_fused_fp8_kv():
    import flashinfer_amd as fi

    torch.manual_seed(3)
    Hq, Hkv, D, page = 32, 8, 128, 16
    kv_lens = [1024, 511]
    indptr, indices, lpl, kc, vc = _paged(2, kv_lens, Hkv, D, page)
    kc8 = kc.clamp(-8, 8).to(torch.float8_e4m3fn)
    vc8 = vc.clamp(-8, 8).to(torch.float8_e4m3fn)
    q = torch.randn(2, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16, kv_data_type=torch.float8_e4m3fn)
    assert w._fused
    out = w.run(q, (kc8, vc8))
    for b in range(2):
        kk, vv = _gather(indptr, indices, kv_lens,
                         kc8.to(torch.bfloat16), vc8.to(torch.bfloat16), b, page)
        ref = sdpa_ref(q[b], kk, vv)
        torch.testing.assert_close(out[b].float(), ref, atol=6e-2, rtol=6e-2)


def test_decode_mfma_fp8_kv():
    """fp8 e4m3 KV cache on the GQA-8 MFMA decode route (dequant happens in
    the LDS staging write; k_scale folds into sm_scale, v_scale on the out)."""
    import flashinfer_amd as fi

    torch.manual_seed(5)
    Hq, Hkv, D, page = 64, 8, 128, 16
    # 8192 > _MFMA_MAX_KV: fp8 KV stays on the MFMA route at any length
    # (no tc fallback exists for fp8; see decode._MFMA_MAX_KV_F8)
    kv_lens = [8192, 511, 77]
    indptr, indices, lpl, kc, vc = _paged(3, kv_lens, Hkv, D, page)
    kc8 = kc.clamp(-8, 8).to(torch.float8_e4m3fn)
    vc8 = vc.clamp(-8, 8).to(torch.float8_e4m3fn)
    q = torch.randn(3, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16, kv_data_type=torch.float8_e4m3fn)
    assert w._fused_mfma, "GQA-8 fp8-KV must take the MFMA decode route"
    out = w.run(q, (kc8, vc8), k_scale=0.5, v_scale=2.0)
    for b in range(3):
        kk, vv = _gather(indptr, indices, kv_lens,
                         kc8.to(torch.bfloat16), vc8.to(torch.bfloat16), b, page)
        ref = sdpa_ref(q[b], kk, vv, sm_scale=0.5 / D ** 0.5) * 2.0
        torch.testing.assert_close(out[b].float(), ref, atol=6e-2, rtol=6e-2)


@pytest.mark.parametrize("Hq,Hkv,kv_dtype", [
    (48, 8, torch.bfloat16),    # GROUP 6 (Mixtral-8x22B shape)
    (56, 8, torch.bfloat16),    # GROUP 7 (Yi-34B shape)
    (32, 32, torch.float8_e4m3fn),   # MHA GROUP 1, fp8 KV
    (32, 8, torch.float8_e4m3fn),    # GROUP 4, fp8 KV (mfma beats fused fp8)
    (32, 8, torch.bfloat16),         # GROUP 4 bf16, kv>2048: past fused range
])
def test_decode_mfma_small_groups(Hq, Hkv, kv_dtype):
    """Non-power-of-2 / small GQA groups on the MFMA route: the 32x32 tile's
    q dim is zero-padded, so any group <= 32 rides the same kernel."""
    import flashinfer_amd as fi

    torch.manual_seed(7)
    D, page = 128, 16
    kv_lens = [3000, 511, 64]
    indptr, indices, lpl, kc, vc = _paged(3, kv_lens, Hkv, D, page)
    if kv_dtype != torch.bfloat16:
        kc = kc.clamp(-8, 8).to(kv_dtype)
        vc = vc.clamp(-8, 8).to(kv_dtype)
    q = torch.randn(3, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16, kv_data_type=kv_dtype)
    assert w._fused_mfma, f"group {Hq // Hkv} must take the MFMA route here"
    out = w.run(q, (kc, vc))
    kvb = kc.to(torch.bfloat16) if kv_dtype != torch.bfloat16 else kc
    vvb = vc.to(torch.bfloat16) if kv_dtype != torch.bfloat16 else vc
    for b in range(3):
        kk, vv = _gather(indptr, indices, kv_lens, kvb, vvb, b, page)
        ref = sdpa_ref(q[b], kk, vv)
        torch.testing.assert_close(out[b].float(), ref, atol=6e-2, rtol=6e-2)


def test_decode_odd_group_tc_fallback():
    """Groups outside the instantiated vector/mfma sets (e.g. 3) must route
    to the group-agnostic prefill-based tc path, not fail at dispatch."""
    import flashinfer_amd as fi

    torch.manual_seed(9)
    Hq, Hkv, D, page = 24, 8, 128, 16   # group 3
    kv_lens = [700, 33]
    indptr, indices, lpl, kc, vc = _paged(2, kv_lens, Hkv, D, page)
    q = torch.randn(2, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16)
    assert w._tc, "group 3 must fall back to the tensor-core path"
    out = w.run(q, (kc, vc))
    for b in range(2):
        kk, vv = _gather(indptr, indices, kv_lens, kc, vc, b, page)
        torch.testing.assert_close(out[b].float(), sdpa_ref(q[b], kk, vv),
                                   atol=3e-2, rtol=3e-2)


def test_decode_tensor_cores_opt_out():
    """Explicit use_tensor_cores=False must never route to the prefill-MFMA
    path, and explicit True must (advisor r01 contract fix)."""
    import flashinfer_amd as fi

    Hq, Hkv, D, page = 64, 8, 128, 16
    kv_lens = [8192] * 4  # long kv: auto would pick tc at group 8
    indptr, indices, lpl, kc, vc = _paged(4, kv_lens, Hkv, D, page)
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", use_tensor_cores=False)
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert not w._tc and not w._fused
    w2 = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", use_tensor_cores=True)
    w2.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w2._tc
    w3 = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w3.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w3._tc  # auto at group 8 / long kv
    q = torch.randn(4, Hq, D, dtype=torch.bfloat16, device="cuda")
    o1 = w.run(q, (kc, vc))
    o2 = w2.run(q, (kc, vc))
    torch.testing.assert_close(o1.float(), o2.float(), atol=2e-2, rtol=2e-2)


def test_decode_strided_q_from_fused_qkv():
    """q sliced out of a fused QKV projection buffer (non-contiguous row
    stride) must work on every decode route — the kernels take
    q_stride_n/h, not an implicit contiguity assumption."""
    import flashinfer_amd as fi

    torch.manual_seed(8)
    Hq, Hkv, D, page = 64, 8, 128, 16
    kv_lens = [1024, 333]
    indptr, indices, lpl, kc, vc = _paged(2, kv_lens, Hkv, D, page)
    qkv = torch.randn(2, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16,
                      device="cuda")
    q = qkv[:, : Hq * D].view(2, Hq, D)   # row stride = (Hq+2*Hkv)*D
    assert not q.is_contiguous()
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16)
    assert w._fused_mfma
    out = w.run(q, (kc, vc))
    ref_out = w.run(q.contiguous(), (kc, vc))
    assert torch.equal(out, ref_out)
    # vector route too
    w2 = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD",
                                               use_tensor_cores=False)
    w2.plan(indptr, indices, lpl, Hq, Hkv, D, page,
            q_data_type=torch.bfloat16)
    out2 = w2.run(q, (kc, vc))
    torch.testing.assert_close(out2.float(), ref_out.float(), atol=3e-2,
                               rtol=3e-2)


def test_decode_fp16_routes():
    """fp16 instantiations of the MFMA and fused decode kernels (everything
    else in the suite runs bf16)."""
    import flashinfer_amd as fi

    torch.manual_seed(10)
    D, page = 128, 16
    kv_lens = [777, 128]
    for Hq, Hkv, utc, want_mfma in ((64, 8, None, True),
                                    (32, 8, None, True),
                                    (32, 8, False, False)):
        indptr, indices, lpl, kc, vc = _paged(2, kv_lens, Hkv, D, page)
        kc = kc.to(torch.float16)
        vc = vc.to(torch.float16)
        q = torch.randn(2, Hq, D, dtype=torch.float16, device="cuda")
        ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
        w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD",
                                                  use_tensor_cores=utc)
        w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
               q_data_type=torch.float16, kv_data_type=torch.float16)
        assert w._fused_mfma == want_mfma
        out = w.run(q, (kc, vc))
        for b in range(2):
            kk, vv = _gather(indptr, indices, kv_lens, kc, vc, b, page)
            ref = sdpa_ref(q[b], kk, vv)
            torch.testing.assert_close(out[b].float(), ref, atol=3e-2,
                                       rtol=3e-2)
