"""GPU tests: packed custom masks, holistic BatchAttention, hipGraph capture."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_single_prefill_custom_mask():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    qo, kv, Hq, Hkv, D = 64, 100, 8, 2, 128
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    mask = torch.rand(qo, kv, device="cuda") > 0.3
    mask[:, 0] = True
    out = fi.single_prefill_with_kv_cache(q, k, v, custom_mask=mask)
    g = Hq // Hkv
    kf = k.float().repeat_interleave(g, 1)
    vf = v.float().repeat_interleave(g, 1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), kf) / math.sqrt(D)
    logits = logits.masked_fill(~mask[None], float("-inf"))
    ref = torch.einsum("hml,lhd->mhd", torch.softmax(logits, -1), vf)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_batch_attention_mixed():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page = 32, 8, 128, 16
    qo_lens = [1, 64, 1, 17]   # mixed decode + prefill
    kv_lens = [400, 64, 33, 17]
    batch = len(qo_lens)
    pages_per = [(L + page - 1) // page for L in kv_lens]
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                             dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    nnz = sum(qo_lens)
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ba = fi.BatchAttention()
    ba.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page,
            causal=True)
    out, lse = ba.run(q, (k_cache, v_cache))
    assert out.shape == q.shape and out.isfinite().all() and lse.isfinite().all()


def test_decode_hipgraph_capture():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page, bs, kv = 32, 8, 128, 16, 8, 256
    pages_per = kv // page
    indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32,
                          device="cuda")
    npages = bs * pages_per
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", use_cuda_graph=True)
    w.plan(indptr, indices, last, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    # warmup on a side stream then capture
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            w.run(q, (kc, vc), out=out)
    torch.cuda.current_stream().wait_stream(s)
    eager = w.run(q, (kc, vc)).clone()
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        w.run(q, (kc, vc), out=out)
    out.zero_()
    graph.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out, eager)
    # new inputs, replay again
    q.copy_(torch.randn_like(q))
    eager2 = w.run(q, (kc, vc)).clone()
    graph.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out, eager2)


def test_cu_mask_streams():
    import flashinfer_amd as fi

    streams, counts = fi.split_device_cu_streams(torch.device("cuda:0"), [192, 32])
    assert counts == [192, 32]
    x = torch.randn(1024, 1024, device="cuda")
    with torch.cuda.stream(streams[1]):
        y = x @ x
    torch.cuda.synchronize()
    assert y.isfinite().all()


def test_prefill_hipgraph_capture():
    """Prefill run() captures into a hipGraph (fixed grid, no alloc)."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    bs, s = 4, 256
    pages_per = s // page
    npages = bs * pages_per
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs * s, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    kv_indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                             device="cuda")
    kv_indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, lpl, Hq, Hkv, D, page, causal=True)
    w.run(q, (kc, vc), out=out)  # warm
    torch.cuda.synchronize()
    ref = out.clone()

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        w.run(q, (kc, vc), out=out)
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref.float())
    # replay with mutated inputs picks up the new data
    q.normal_()
    w.run(q, (kc, vc), out=ref)
    torch.cuda.synchronize()
    ref2 = ref.clone()
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref2.float())


def test_mla_hipgraph_capture():
    import flashinfer_amd as fi

    torch.manual_seed(1)
    B, H, page = 4, 16, 16
    kv = 256
    pages_per = kv // page
    npages = B * pages_per
    ckv = torch.randn(npages, page, 512, dtype=torch.bfloat16, device="cuda") / 4
    kpe = torch.randn(npages, page, 64, dtype=torch.bfloat16, device="cuda") / 4
    qn = torch.randn(B, H, 512, dtype=torch.bfloat16, device="cuda") / 4
    qp = torch.randn(B, H, 64, dtype=torch.bfloat16, device="cuda") / 4
    qo_indptr = torch.arange(0, B + 1, dtype=torch.int32, device="cuda")
    kv_indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                             device="cuda")
    kv_indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    kvlen = torch.full((B,), kv, dtype=torch.int32, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    w.plan(qo_indptr, kv_indptr, kv_indices, kvlen, H, 512, 64, page,
           causal=False, sm_scale=576 ** -0.5)
    out = torch.empty(B, H, 512, dtype=torch.bfloat16, device="cuda")
    w.run(qn, qp, ckv, kpe, out=out)
    torch.cuda.synchronize()
    ref = out.clone()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        w.run(qn, qp, ckv, kpe, out=out)
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref.float())


def test_tensor_core_decode_graph_capture():
    """GQA-8 decode (auto tensor-core route) captures into a hipGraph."""
    import flashinfer_amd as fi

    torch.manual_seed(4)
    Hq, Hkv, D, page = 64, 8, 128, 16
    bs, pages_per = 8, 8
    npages = bs * pages_per
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                          device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", use_tensor_cores=True)
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w._tc  # explicit opt-in must pin the tensor-core route
    w.run(q, (kc, vc), out=out)
    torch.cuda.synchronize()
    ref = out.clone()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        w.run(q, (kc, vc), out=out)
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref.float())


def test_fused_decode_graph_capture():
    """Short-kv decode (fused whole-request route) captures into a hipGraph."""
    import flashinfer_amd as fi

    torch.manual_seed(5)
    Hq, Hkv, D, page = 64, 8, 128, 16
    bs, pages_per = 8, 8
    npages = bs * pages_per
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    indptr = torch.arange(0, npages + 1, pages_per, dtype=torch.int32,
                          device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w._fused
    w.run(q, (kc, vc), out=out)
    torch.cuda.synchronize()
    ref = out.clone()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        w.run(q, (kc, vc), out=out)
    out.zero_()
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(out, ref)


def test_mfma_split_inkernel_merge_graph_capture():
    """MFMA decode with cross-WG split + same-XCD in-kernel merge captures
    into a hipGraph and replays bitwise — the arrival counters self-reset,
    so no per-replay zeroing is needed."""
    import flashinfer_amd as fi

    torch.manual_seed(6)
    Hq, Hkv, D, page = 64, 8, 128, 16
    bs, L = 16, 1024
    npages = bs * (L // page)
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    indptr = torch.arange(0, bs + 1, dtype=torch.int32, device="cuda") * (L // page)
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert w._fused_mfma and w._mfma_split > 1
    assert w._mfma_counters is not None, "in-kernel merge must be active"
    w.run(q, (kc, vc), out=out)
    torch.cuda.synchronize()
    ref = out.clone()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        w.run(q, (kc, vc), out=out)
    for _ in range(3):
        out.zero_()
        g.replay()
        torch.cuda.synchronize()
        torch.testing.assert_close(out, ref)
    assert int(w._mfma_counters.abs().sum().item()) == 0, "counters must self-reset"
