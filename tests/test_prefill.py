"""GPU numerics: prefill attention (MFMA kernel) vs fp32 PyTorch reference."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def ref_attn(q, k, v, causal=False, sm_scale=None, window_left=-1, soft_cap=0.0):
    # q [M, Hq, D], k/v [L, Hkv, D] -> [M, Hq, D] fp32
    M, Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    qf = q.float().transpose(0, 1)  # [Hq, M, D]
    kf = k.float().repeat_interleave(g, dim=1).transpose(0, 1)  # [Hq, L, D]
    vf = v.float().repeat_interleave(g, dim=1).transpose(0, 1)
    scale = sm_scale if sm_scale is not None else 1 / math.sqrt(D)
    logits = qf @ kf.transpose(-1, -2) * scale  # [Hq, M, L]
    if soft_cap > 0:
        logits = soft_cap * torch.tanh(logits / soft_cap)
    qpos = torch.arange(M, device=q.device)[:, None]
    kpos = torch.arange(L, device=q.device)[None, :]
    diag = L - M
    if causal:
        logits = logits.masked_fill((kpos > qpos + diag)[None], float("-inf"))
    if window_left >= 0:
        logits = logits.masked_fill((kpos < qpos + diag - window_left)[None],
                                    float("-inf"))
    p = torch.softmax(logits, dim=-1)
    return (p @ vf).transpose(0, 1)


@pytest.mark.parametrize("qo_len,kv_len", [(1, 1), (17, 17), (128, 128),
                                           (333, 500), (1024, 1024), (64, 2048)])
@pytest.mark.parametrize("Hq,Hkv", [(32, 8), (8, 8), (16, 1)])
@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("D", [128])
def test_single_prefill(qo_len, kv_len, Hq, Hkv, causal, D):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    q = torch.randn(qo_len, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    out, lse = fi.single_prefill_with_kv_cache(q, k, v, causal=causal,
                                               return_lse=True)
    ref = ref_attn(q, k, v, causal=causal)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    assert lse.isfinite().all()


@pytest.mark.parametrize("D", [64, 256])
def test_single_prefill_head_dims(D):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    q = torch.randn(200, 8, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(200, 2, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(200, 2, D, dtype=torch.bfloat16, device="cuda")
    out = fi.single_prefill_with_kv_cache(q, k, v, causal=True)
    ref = ref_attn(q, k, v, causal=True)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_single_prefill_window_softcap():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    q = torch.randn(256, 8, 128, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(300, 2, 128, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(300, 2, 128, dtype=torch.bfloat16, device="cuda")
    out = fi.single_prefill_with_kv_cache(q, k, v, causal=True, window_left=64,
                                          logits_soft_cap=20.0)
    ref = ref_attn(q, k, v, causal=True, window_left=64, soft_cap=20.0)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("page_size", [1, 16])
@pytest.mark.parametrize("causal", [False, True])
def test_batch_prefill_paged(page_size, causal):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D = 32, 8, 128
    qo_lens = [3, 128, 77, 1]
    kv_lens = [10, 128, 200, 1]
    batch = len(qo_lens)
    nnz_q = sum(qo_lens)
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    pages_per = [(L + page_size - 1) // page_size for L in kv_lens]
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                             dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last_page_len = torch.tensor([(L - 1) % page_size + 1 for L in kv_lens],
                                 dtype=torch.int32, device="cuda")
    k_cache = torch.randn(npages, page_size, Hkv, D, dtype=torch.bfloat16,
                          device="cuda")
    v_cache = torch.randn(npages, page_size, Hkv, D, dtype=torch.bfloat16,
                          device="cuda")
    q = torch.randn(nnz_q, Hq, D, dtype=torch.bfloat16, device="cuda")

    ws = torch.empty(16 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    wrapper = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    wrapper.plan(qo_indptr, kv_indptr, kv_indices, last_page_len, Hq, Hkv, D,
                 page_size, causal=causal)
    out = wrapper.run(q, (k_cache, v_cache))

    for b in range(batch):
        L = kv_lens[b]
        toks_k, toks_v = [], []
        for pi in range(pages_per[b]):
            page = int(kv_indices[int(kv_indptr[b]) + pi])
            n = min(page_size, L - pi * page_size)
            toks_k.append(k_cache[page, :n])
            toks_v.append(v_cache[page, :n])
        kk, vv = torch.cat(toks_k), torch.cat(toks_v)
        qb = q[int(qo_indptr[b]) : int(qo_indptr[b + 1])]
        ref = ref_attn(qb, kk, vv, causal=causal)
        torch.testing.assert_close(
            out[int(qo_indptr[b]) : int(qo_indptr[b + 1])].float(), ref,
            atol=3e-2, rtol=3e-2, msg=f"req {b}",
        )


def test_batch_prefill_ragged():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D = 32, 8, 128
    qo_lens = [64, 1000, 1]
    kv_lens = qo_lens
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    kv_indptr = qo_indptr.clone()
    nnz = sum(qo_lens)
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(16 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    wrapper = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    wrapper.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=True)
    out = wrapper.run(q, k, v)
    for b in range(len(qo_lens)):
        s, e = int(qo_indptr[b]), int(qo_indptr[b + 1])
        ref = ref_attn(q[s:e], k[s:e], v[s:e], causal=True)
        torch.testing.assert_close(out[s:e].float(), ref, atol=3e-2, rtol=3e-2)


def test_prefill_with_empty_request():
    """qo_len == 0 requests in the batch are skipped cleanly."""
    import flashinfer_amd as fi

    torch.manual_seed(1)
    Hq, Hkv, D, page = 8, 2, 128, 16
    qo_indptr = torch.tensor([0, 32, 32, 64], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, 2, 2, 4], dtype=torch.int32, device="cuda")
    kv_indices = torch.arange(4, dtype=torch.int32, device="cuda")
    lpl = torch.tensor([page, 0, page], dtype=torch.int32, device="cuda")
    kc = torch.randn(4, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(4, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(64, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(16 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, lpl, Hq, Hkv, D, page, causal=True)
    out = w.run(q, (kc, vc))
    assert out.isfinite().all()
    # spot-check request 2 against a dense reference
    import math
    kv = kc.view(-1, Hkv, D)[32:64].float().repeat_interleave(Hq // Hkv, 1)
    vv = vc.view(-1, Hkv, D)[32:64].float().repeat_interleave(Hq // Hkv, 1)
    logits = torch.einsum("mhd,lhd->hml", q[32:].float(), kv) / math.sqrt(D)
    mask = torch.arange(32, device="cuda")[None, :] > torch.arange(
        32, device="cuda")[:, None]
    logits = logits.masked_fill(mask[None], float("-inf"))
    ref = torch.einsum("hml,lhd->mhd", torch.softmax(logits, -1), vv)
    torch.testing.assert_close(out[32:].float(), ref, atol=3e-2, rtol=3e-2)


# ------------- head_dim_qk 192 / head_dim_vo 128 (DeepSeek MHA) -------------

def ref_attn_vo(q, k, v, causal=False, sm_scale=None):
    """Reference with head_dim_vo != head_dim_qk: v [L, Hkv, Dvo]."""
    M, Hq, D = q.shape
    L, Hkv, _ = k.shape
    g = Hq // Hkv
    qf = q.float().transpose(0, 1)
    kf = k.float().repeat_interleave(g, dim=1).transpose(0, 1)
    vf = v.float().repeat_interleave(g, dim=1).transpose(0, 1)
    scale = sm_scale if sm_scale is not None else 1 / math.sqrt(D)
    logits = qf @ kf.transpose(-1, -2) * scale
    if causal:
        qpos = torch.arange(M, device=q.device)[:, None]
        kpos = torch.arange(L, device=q.device)[None, :]
        logits = logits.masked_fill((kpos > qpos + (L - M))[None], float("-inf"))
    p = torch.softmax(logits, dim=-1)
    return (p @ vf).transpose(0, 1)


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("Hq,Hkv", [(16, 16), (16, 4)])
def test_ragged_prefill_hd192_128(causal, Hq, Hkv):
    """DeepSeek MHA shape (192 qk / 128 vo) on the ragged path — the
    BASELINE.md ragged-prefill row config (reference
    benchmarks/samples/sample_testlist_output.csv:5)."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    qo_lens = [128, 333, 64]
    kv_lens = [128, 333, 517]
    nnz_q, nnz_kv = sum(qo_lens), sum(kv_lens)
    q = torch.randn(nnz_q, Hq, 192, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz_kv, Hkv, 192, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(nnz_kv, Hkv, 128, dtype=torch.bfloat16, device="cuda")
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(kv_lens), 0)),
                             dtype=torch.int32, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, 192, head_dim_vo=128, causal=causal,
           q_data_type=torch.bfloat16)
    out = w.run(q, k, v)
    assert out.shape == (nnz_q, Hq, 128)
    for b in range(3):
        qs, qe = int(qo_indptr[b]), int(qo_indptr[b + 1])
        ks, ke = int(kv_indptr[b]), int(kv_indptr[b + 1])
        ref = ref_attn_vo(q[qs:qe], k[ks:ke], v[ks:ke], causal=causal)
        torch.testing.assert_close(out[qs:qe].float(), ref, atol=3e-2, rtol=3e-2)


# ---------------------------- prefill split-KV ----------------------------

@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("qo_len,kv_len", [(16, 8192), (1, 16384), (128, 4096)])
def test_prefill_split_kv_ragged(causal, qo_len, kv_len):
    """Short-q/long-kv prefill must split KV across workgroups and merge
    (reference scheduler.cuh:545 role) — and match the fp32 reference."""
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D = 8, 8, 128
    q = torch.randn(qo_len, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device="cuda")
    qo_indptr = torch.tensor([0, qo_len], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, kv_len], dtype=torch.int32, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=causal,
           q_data_type=torch.bfloat16)
    assert w._split, "short-q/long-kv plan must split KV"
    out, lse = w.run(q, k, v, return_lse=True)
    ref = ref_attn(q, k, v, causal=causal)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    # lse sanity (non-causal row 0 vs torch)
    kf = k.float()
    logits = torch.einsum("hd,lhd->hl", q[0].float(), kf) / math.sqrt(D)
    if causal:
        kpos = torch.arange(kv_len, device=q.device)
        logits = logits.masked_fill((kpos > (kv_len - qo_len))[None],
                                    float("-inf"))
    ref_lse = torch.logsumexp(logits, -1) / math.log(2)
    torch.testing.assert_close(lse[0], ref_lse, atol=2e-2, rtol=2e-2)


def test_prefill_split_kv_paged():
    import flashinfer_amd as fi

    torch.manual_seed(1)
    Hq, Hkv, D, page = 32, 8, 128, 16
    qo_lens = [4, 2]
    kv_lens = [6000, 3111]
    pages_per = [(L + page - 1) // page for L in kv_lens]
    qo_indptr = torch.tensor([0, 4, 6], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                             dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    lpl = torch.tensor([(L - 1) % page + 1 for L in kv_lens], dtype=torch.int32,
                       device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(6, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, lpl, Hq, Hkv, D, page,
           causal=True, q_data_type=torch.bfloat16)
    assert w._split
    out = w.run(q, (kc, vc))
    for b in range(2):
        qs, qe = int(qo_indptr[b]), int(qo_indptr[b + 1])
        toks_k, toks_v = [], []
        base = [0, pages_per[0]][b]
        for p_ in range(pages_per[b]):
            pg = int(kv_indices[base + p_])
            n = min(page, kv_lens[b] - p_ * page)
            toks_k.append(kc[pg, :n])
            toks_v.append(vc[pg, :n])
        kk = torch.cat(toks_k, 0)
        vv = torch.cat(toks_v, 0)
        ref = ref_attn(q[qs:qe], kk, vv, causal=True)
        torch.testing.assert_close(out[qs:qe].float(), ref, atol=3e-2, rtol=3e-2)


def test_prefill_strided_q_from_fused_qkv():
    """Ragged prefill with q sliced from a fused QKV buffer (non-contiguous
    row stride)."""
    import flashinfer_amd as fi

    torch.manual_seed(9)
    nnz, Hq, Hkv, D = 512, 32, 8, 128
    qkv = torch.randn(nnz, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16,
                      device="cuda")
    q = qkv[:, : Hq * D].view(nnz, Hq, D)
    assert not q.is_contiguous()
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    qo_indptr = torch.tensor([0, 256, nnz], dtype=torch.int32, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, qo_indptr, Hq, Hkv, D, causal=True,
           q_data_type=torch.bfloat16)
    out = w.run(q, k, v)
    ref = w.run(q.contiguous(), k, v)
    assert torch.equal(out, ref)


def test_prefill_fp16():
    import flashinfer_amd as fi

    torch.manual_seed(11)
    nnz, Hq, Hkv, D = 384, 32, 8, 128
    q = torch.randn(nnz, Hq, D, dtype=torch.float16, device="cuda")
    k = torch.randn(nnz, Hkv, D, dtype=torch.float16, device="cuda")
    v = torch.randn(nnz, Hkv, D, dtype=torch.float16, device="cuda")
    qo_indptr = torch.tensor([0, nnz], dtype=torch.int32, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, qo_indptr, Hq, Hkv, D, causal=True,
           q_data_type=torch.float16)
    out = w.run(q, k, v)
    ref = ref_attn(q, k, v, causal=True)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
