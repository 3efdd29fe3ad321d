"""GPU: persistent holistic BatchAttention (mixed prefill+decode, one launch)
vs fp32 reference; parity with the reference BatchAttention contract
(flashinfer/attention/_core.py:44)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _mixed_batch(qo_lens, kv_lens, Hkv, D, page, dtype=torch.bfloat16):
    pages_per = [(L + page - 1) // page for L in kv_lens]
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                             dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    return qo_indptr, kv_indptr, kv_indices, kv_len_arr, kc, vc, pages_per


def _ref(q, kk, vv, causal, qo_len, kv_len):
    Hq, D = q.shape[1], q.shape[2]
    g = Hq // kk.shape[1]
    qf = q.float().transpose(0, 1)
    kf = kk.float().repeat_interleave(g, dim=1).transpose(0, 1)
    vf = vv.float().repeat_interleave(g, dim=1).transpose(0, 1)
    logits = qf @ kf.transpose(-1, -2) / math.sqrt(D)
    if causal:
        qpos = torch.arange(qo_len, device=q.device)[:, None]
        kpos = torch.arange(kv_len, device=q.device)[None, :]
        logits = logits.masked_fill((kpos > qpos + (kv_len - qo_len))[None],
                                    float("-inf"))
    return (torch.softmax(logits, dim=-1) @ vf).transpose(0, 1)


@pytest.mark.parametrize("Hq,Hkv,D", [(64, 8, 128), (32, 8, 128), (16, 4, 64)])
def test_holistic_mixed_batch(Hq, Hkv, D):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    page = 16
    # 50/50 mixed: prefill requests + single-token decode requests
    qo_lens = [512, 1, 1, 300, 1, 128, 1, 1]
    kv_lens = [512, 1024, 777, 300, 2048, 128, 64, 1500]
    (qo_indptr, kv_indptr, kv_indices, kv_len_arr, kc, vc,
     pages_per) = _mixed_batch(qo_lens, kv_lens, Hkv, D, page)
    nnz = sum(qo_lens)
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page,
           causal=True, q_data_type=torch.bfloat16)
    assert w._persistent, "eligible mixed batch must use the persistent kernel"
    out, lse = w.run(q, (kc, vc))
    assert out.shape == (nnz, Hq, D)
    for b in range(len(qo_lens)):
        qs, qe = int(qo_indptr[b]), int(qo_indptr[b + 1])
        base = int(kv_indptr[b])
        toks_k, toks_v = [], []
        for p_ in range(pages_per[b]):
            pg = int(kv_indices[base + p_])
            n = min(page, kv_lens[b] - p_ * page)
            toks_k.append(kc[pg, :n])
            toks_v.append(vc[pg, :n])
        kk = torch.cat(toks_k, 0)
        vv = torch.cat(toks_v, 0)
        ref = _ref(q[qs:qe], kk, vv, True, qo_lens[b], kv_lens[b])
        torch.testing.assert_close(out[qs:qe].float(), ref, atol=3e-2,
                                   rtol=3e-2,
                                   msg=lambda m: f"req {b}: {m}")


def test_holistic_deterministic():
    """Atomic ticket order must not affect results (disjoint outputs)."""
    import flashinfer_amd as fi

    torch.manual_seed(1)
    Hq, Hkv, D, page = 64, 8, 128, 16
    qo_lens = [256, 1, 1, 1, 64]
    kv_lens = [256, 512, 2048, 96, 64]
    (qo_indptr, kv_indptr, kv_indices, kv_len_arr, kc, vc,
     _) = _mixed_batch(qo_lens, kv_lens, Hkv, D, page)
    q = torch.randn(sum(qo_lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page,
           causal=True, q_data_type=torch.bfloat16)
    assert w._persistent
    o1, _ = w.run(q, (kc, vc))
    o2, _ = w.run(q, (kc, vc))
    assert torch.equal(o1, o2)


def test_holistic_fp16():
    """fp16 instantiation of the persistent holistic kernel (exercises the
    dtype-conditional P-pack through the shared prefill/decode bodies)."""
    import flashinfer_amd as fi

    torch.manual_seed(1)
    Hq, Hkv, D, page = 64, 8, 128, 16
    qo_lens = [256, 1, 1, 128]
    kv_lens = [256, 1024, 512, 128]
    (qo_indptr, kv_indptr, kv_indices, kv_len_arr, kc, vc,
     pages_per) = _mixed_batch(qo_lens, kv_lens, Hkv, D, page)
    kc = kc.to(torch.float16)
    vc = vc.to(torch.float16)
    nnz = sum(qo_lens)
    q = torch.randn(nnz, Hq, D, dtype=torch.float16, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page,
           causal=True, q_data_type=torch.float16)
    out, lse = w.run(q, (kc, vc))
    for b in range(len(qo_lens)):
        qs, qe = int(qo_indptr[b]), int(qo_indptr[b + 1])
        base = int(kv_indptr[b])
        toks_k, toks_v = [], []
        for p_ in range(pages_per[b]):
            pg = int(kv_indices[base + p_])
            n = min(page, kv_lens[b] - p_ * page)
            toks_k.append(kc[pg, :n])
            toks_v.append(vc[pg, :n])
        ref = _ref(q[qs:qe], torch.cat(toks_k, 0), torch.cat(toks_v, 0),
                   True, qo_lens[b], kv_lens[b])
        torch.testing.assert_close(out[qs:qe].float(), ref, atol=3e-2,
                                   rtol=3e-2, msg=lambda m: f"req {b}: {m}")
