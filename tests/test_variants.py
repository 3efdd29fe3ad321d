"""GPU tests: ALiBi position bias, attention sinks, deterministic split-KV."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _alibi_slopes(Hq, device):
    i = torch.arange(1, Hq + 1, device=device, dtype=torch.float32)
    return torch.exp2(-8.0 * i / Hq)


def test_decode_alibi():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, kv = 8, 2, 128, 500
    q = torch.randn(Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    out = fi.single_decode_with_kv_cache(q, k, v, pos_encoding_mode="ALIBI")
    g = Hq // Hkv
    kf = k.float().repeat_interleave(g, 1)
    vf = v.float().repeat_interleave(g, 1)
    logits = torch.einsum("hd,lhd->hl", q.float(), kf) / math.sqrt(D)
    slopes = _alibi_slopes(Hq, "cuda")
    dist = (kv - 1) - torch.arange(kv, device="cuda", dtype=torch.float32)
    logits = logits - slopes[:, None] * dist[None, :]
    ref = torch.einsum("hl,lhd->hd", torch.softmax(logits, -1), vf)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_prefill_alibi():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    qo, kv, Hq, Hkv, D = 128, 128, 8, 2, 128
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    out = fi.single_prefill_with_kv_cache(q, k, v, causal=True,
                                          pos_encoding_mode="ALIBI")
    g = Hq // Hkv
    kf = k.float().repeat_interleave(g, 1)
    vf = v.float().repeat_interleave(g, 1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), kf) / math.sqrt(D)
    slopes = _alibi_slopes(Hq, "cuda")
    qp = torch.arange(qo, device="cuda", dtype=torch.float32)[:, None]
    kp = torch.arange(kv, device="cuda", dtype=torch.float32)[None, :]
    logits = logits - slopes[:, None, None] * (qp - kp)
    logits = logits.masked_fill((kp > qp)[None], float("-inf"))
    ref = torch.einsum("hml,lhd->mhd", torch.softmax(logits, -1), vf)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_decode_sinks():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, kv, page = 8, 2, 128, 300, 16
    bs = 3
    pp = (kv + page - 1) // page
    indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    npages = bs * pp
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    sinks = torch.randn(Hq, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, last, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = w.run(q, (kc, vc), sinks=sinks)
    # reference with virtual sink logit in the denominator
    g = Hq // Hkv
    for b in range(bs):
        rows_k = kc.view(-1, Hkv, D)[b * pp * page : b * pp * page + kv]
        rows_v = vc.view(-1, Hkv, D)[b * pp * page : b * pp * page + kv]
        kf = rows_k.float().repeat_interleave(g, 1)
        vf = rows_v.float().repeat_interleave(g, 1)
        logits = torch.einsum("hd,lhd->hl", q[b].float(), kf) / math.sqrt(D)
        full = torch.cat([logits, sinks[:, None]], dim=1)
        p = torch.softmax(full, -1)[:, :-1]
        ref = torch.einsum("hl,lhd->hd", p, vf)
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)


def test_decode_batch_invariance():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, page = 32, 8, 128, 16
    kv_lens = [1024, 2048]
    kc = torch.randn(400, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(400, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(2, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")

    def run(batch_idx):
        lens = [kv_lens[i] for i in batch_idx]
        pp = [(L + page - 1) // page for L in lens]
        indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pp), 0)),
                              dtype=torch.int32, device="cuda")
        # request i uses pages [i*128, ...)
        indices = torch.cat([
            torch.arange(i * 128, i * 128 + n, dtype=torch.int32, device="cuda")
            for i, n in zip(batch_idx, pp)
        ])
        last = torch.tensor([(L - 1) % page + 1 for L in lens], dtype=torch.int32,
                            device="cuda")
        w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
        w.plan(indptr, indices, last, Hq, Hkv, D, page,
               q_data_type=torch.bfloat16, fixed_split_size=512)
        return w.run(q[list(batch_idx)], (kc, vc))

    both = run([0, 1])
    solo0 = run([0])
    solo1 = run([1])
    # fixed split size -> identical partition -> bitwise identical results
    assert torch.equal(both[0], solo0[0])
    assert torch.equal(both[1], solo1[0])


def test_decode_fp8_kv_cache():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    Hq, Hkv, D, kv, page, bs = 32, 8, 128, 1024, 16, 4
    pp = kv // page
    indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    npages = bs * pp
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    k_scale, v_scale = 0.05, 0.04
    kc8 = (torch.randn(npages, page, Hkv, D, device="cuda") * 8).to(
        torch.float8_e4m3fn)
    vc8 = (torch.randn(npages, page, Hkv, D, device="cuda") * 8).to(
        torch.float8_e4m3fn)
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, last, Hq, Hkv, D, page, q_data_type=torch.bfloat16,
           kv_data_type=torch.float8_e4m3fn)
    out = w.run(q, (kc8, vc8), k_scale=k_scale, v_scale=v_scale)
    # reference on dequantized cache
    from tests.test_decode import sdpa_ref

    for b in range(bs):
        rows_k = (kc8.float() * k_scale).view(-1, Hkv, D)[b * kv:(b + 1) * kv]
        rows_v = (vc8.float() * v_scale).view(-1, Hkv, D)[b * kv:(b + 1) * kv]
        ref = sdpa_ref(q[b], rows_k.bfloat16(), rows_v.bfloat16())
        torch.testing.assert_close(out[b].float(), ref, atol=5e-2, rtol=5e-2)


def test_append_fp8_quantize():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    H, D, page = 2, 64, 16
    nnz = 40
    npages = 4
    indptr = torch.tensor([0, 4], dtype=torch.int32, device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    last = torch.tensor([nnz - 3 * page], dtype=torch.int32, device="cuda")
    kc = torch.zeros(npages, page, H, D, dtype=torch.float8_e4m3fn, device="cuda")
    vc = torch.zeros(npages, page, H, D, dtype=torch.float8_e4m3fn, device="cuda")
    k_new = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    v_new = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    append_indptr = torch.tensor([0, nnz], dtype=torch.int32, device="cuda")
    seq_lens = torch.tensor([nnz], dtype=torch.int32, device="cuda")
    bi, pos = fi.get_batch_indices_positions(append_indptr, seq_lens, nnz)
    fi.append_paged_kv_cache(k_new, v_new, bi, pos, (kc, vc), indices, indptr,
                             last, "NHD", k_scale=0.1, v_scale=0.2)
    got_k = kc.view(-1, H, D).float()[:nnz] * 0.1
    torch.testing.assert_close(got_k, k_new.float(), atol=0.05, rtol=0.1)


def test_prefill_fp8_kv_cache():
    import flashinfer_amd as fi
    from tests.test_prefill import ref_attn

    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    qo = kv = 256
    pp = kv // page
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, pp], dtype=torch.int32, device="cuda")
    kv_indices = torch.arange(pp, dtype=torch.int32, device="cuda")
    last = torch.tensor([page], dtype=torch.int32, device="cuda")
    ks, vs = 0.07, 0.06
    kc8 = (torch.randn(pp, page, Hkv, D, device="cuda") * 8).to(torch.float8_e4m3fn)
    vc8 = (torch.randn(pp, page, Hkv, D, device="cuda") * 8).to(torch.float8_e4m3fn)
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, last, Hq, Hkv, D, page, causal=True,
           kv_data_type=torch.float8_e4m3fn)
    out = w.run(q, (kc8, vc8), k_scale=ks, v_scale=vs)
    kd = (kc8.float() * ks).view(kv, Hkv, D).bfloat16()
    vd = (vc8.float() * vs).view(kv, Hkv, D).bfloat16()
    ref = ref_attn(q, kd, vd, causal=True)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_mla_fp8_kv_cache():
    import flashinfer_amd as fi
    from tests.test_mla import mla_ref

    torch.manual_seed(0)
    H, page, kv = 32, 32, 256
    pp = kv // page
    sm = 1.0 / math.sqrt(576)
    kv_indptr = torch.tensor([0, pp], dtype=torch.int32, device="cuda")
    kv_indices = torch.arange(pp, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor([kv], dtype=torch.int32, device="cuda")
    qo_indptr = torch.tensor([0, 1], dtype=torch.int32, device="cuda")
    cs, ps = 0.05, 0.05
    ckv8 = (torch.randn(pp, page, 512, device="cuda") * 8).to(torch.float8_e4m3fn)
    kpe8 = (torch.randn(pp, page, 64, device="cuda") * 8).to(torch.float8_e4m3fn)
    qn = torch.randn(1, H, 512, dtype=torch.bfloat16, device="cuda") / 4
    qp = torch.randn(1, H, 64, dtype=torch.bfloat16, device="cuda") / 4
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, H, 512, 64, page, True,
           sm, torch.bfloat16, kv_data_type=torch.float8_e4m3fn)
    out = w.run(qn, qp, ckv8, kpe8, ckv_scale=cs, kpe_scale=ps)
    ckv_d = (ckv8.float() * cs).view(kv, 512).bfloat16()
    kpe_d = (kpe8.float() * ps).view(kv, 64).bfloat16()
    ref = mla_ref(qn, qp, ckv_d, kpe_d, sm, True, 1, kv)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)
