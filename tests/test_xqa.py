"""GPU: XQA speculative/decode attention vs the underlying wrappers."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_xqa_decode_matches_wrapper():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    B, Hq, Hkv, D, page = 4, 8, 2, 128, 16
    max_pages = 8
    seq_lens = torch.tensor([100, 128, 17, 64], dtype=torch.int32, device="cuda")
    npages = B * max_pages
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    page_table = torch.arange(npages, dtype=torch.int32,
                              device="cuda").view(B, max_pages)
    q = torch.randn(B, 1, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    fi.xqa(q, kc, vc, page_table, seq_lens, out, ws, None, Hkv, page)

    # reference: decode wrapper with the same plan inputs
    lens = seq_lens.long()
    ppr = (lens + page - 1) // page
    kv_indptr = torch.zeros(B + 1, dtype=torch.int32, device="cuda")
    kv_indptr[1:] = ppr.cumsum(0).int()
    kv_indices = torch.cat([page_table[b, :ppr[b]] for b in range(B)]).int()
    last = ((lens - 1) % page + 1).int()
    ws2 = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws2, "NHD")
    w.plan(kv_indptr, kv_indices, last, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16)
    ref = w.run(q.view(B, Hq, D), (kc, vc))
    torch.testing.assert_close(out.view(B, Hq, D).float(), ref.float())


def test_xqa_speculative_matches_prefill():
    import flashinfer_amd as fi

    torch.manual_seed(1)
    B, Hq, Hkv, D, page, nq = 2, 4, 2, 128, 16, 3
    max_pages = 8
    seq_lens = torch.tensor([128, 80], dtype=torch.int32, device="cuda")
    npages = B * max_pages
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    page_table = torch.arange(npages, dtype=torch.int32,
                              device="cuda").view(B, max_pages)
    q = torch.randn(B, 1, nq, Hq, D, dtype=torch.bfloat16, device="cuda")
    out = torch.empty_like(q)
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    fi.xqa(q, kc, vc, page_table, seq_lens, out, ws, None, Hkv, page,
           q_seq_len=nq)
    assert out.isfinite().all()
    # spot-check one request against single_prefill over its gathered KV
    b = 0
    L = int(seq_lens[b])
    rows = kc[page_table[b, : (L + page - 1) // page].long()].view(-1, Hkv, D)[:L]
    vrows = vc[page_table[b, : (L + page - 1) // page].long()].view(-1, Hkv, D)[:L]
    # speculative rows sit at the END of the sequence (right-aligned causal):
    # token i of the nq sees kv[: L - nq + i + 1]
    import math
    for i in range(nq):
        Li = L - nq + i + 1
        logits = torch.einsum(
            "hd,khd->hk", q[b, 0, i].float(),
            rows[:Li].float().repeat_interleave(Hq // Hkv, 1)) / math.sqrt(D)
        ref = torch.einsum("hk,khd->hd", torch.softmax(logits, -1),
                           vrows[:Li].float().repeat_interleave(Hq // Hkv, 1))
        torch.testing.assert_close(out[b, 0, i].float(), ref, atol=3e-2,
                                   rtol=3e-2)


def test_xqa_mla_runs():
    import flashinfer_amd as fi

    torch.manual_seed(2)
    B, Hq, page = 2, 16, 16
    max_pages = 4
    seq_lens = torch.tensor([64, 40], dtype=torch.int32, device="cuda")
    slots = B * max_pages * page
    cache = torch.randn(slots, 576, dtype=torch.bfloat16, device="cuda") / 4
    page_table = torch.arange(B * max_pages, dtype=torch.int32,
                              device="cuda").view(B, max_pages)
    q = torch.randn(B, 1, Hq, 576, dtype=torch.bfloat16, device="cuda") / 4
    out = torch.empty(B, 1, Hq, 512, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(64 << 20, dtype=torch.uint8, device="cuda")
    fi.xqa_mla(q, cache, cache[:, :512], page_table, seq_lens, out, ws, None,
               page)
    assert out.isfinite().all()
    # dense reference with the packed 576-d dot and 512-d output
    import math
    for b in range(B):
        L = int(seq_lens[b])
        rows = cache.view(-1, page, 576)[page_table[b].long()].view(-1, 576)[:L]
        logits = (q[b, 0].float() @ rows.float().t()) / math.sqrt(576)
        ref = torch.softmax(logits, -1) @ rows[:, :512].float()
        torch.testing.assert_close(out[b, 0].float(), ref, atol=3e-2, rtol=3e-2)
