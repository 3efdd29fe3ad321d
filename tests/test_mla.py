"""GPU numerics: DeepSeek MLA paged decode vs dense fp32 reference."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def mla_ref(q_nope, q_pe, ckv, kpe, sm_scale, causal, qo_len, kv_len):
    # q_nope [qo, H, 512], q_pe [qo, H, 64]; ckv [kv, 512], kpe [kv, 64]
    q = torch.cat([q_nope, q_pe], dim=-1).float()  # [qo, H, 576]
    k = torch.cat([ckv, kpe], dim=-1).float()      # [kv, 576]
    logits = torch.einsum("mhd,ld->hml", q, k) * sm_scale
    if causal:
        qpos = torch.arange(qo_len, device=q.device)[:, None]
        kpos = torch.arange(kv_len, device=q.device)[None, :]
        logits = logits.masked_fill((kpos > qpos + kv_len - qo_len)[None],
                                    float("-inf"))
    p = torch.softmax(logits, dim=-1)
    return torch.einsum("hml,ld->mhd", p, ckv.float())


@pytest.mark.parametrize("kv_lens,qo_lens,causal", [
    ([1], [1], False),
    ([64, 129, 1000], [1, 1, 1], True),
    ([2048], [1], True),
    ([33, 80], [2, 4], True),   # incremental prefill (speculative)
    ([1024], [1024], True),     # full chunked MLA prefill
    ([1024, 700], [1024, 300], True),  # mixed full + partial prefill
])
@pytest.mark.parametrize("H", [16, 128])
def test_mla_paged_decode(kv_lens, qo_lens, causal, H):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    page = 32
    batch = len(kv_lens)
    sm_scale = 1.0 / math.sqrt(512 + 64)
    pages_per = [(L + page - 1) // page for L in kv_lens]
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                             dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)),
                             dtype=torch.int32, device="cuda")
    nnz = sum(qo_lens)
    ckv_cache = torch.randn(npages, page, 512, dtype=torch.bfloat16, device="cuda") / 4
    kpe_cache = torch.randn(npages, page, 64, dtype=torch.bfloat16, device="cuda") / 4
    q_nope = torch.randn(nnz, H, 512, dtype=torch.bfloat16, device="cuda") / 4
    q_pe = torch.randn(nnz, H, 64, dtype=torch.bfloat16, device="cuda") / 4

    ws = torch.empty(1024 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, H, 512, 64, page,
           causal, sm_scale, torch.bfloat16)
    out, lse = w.run(q_nope, q_pe, ckv_cache, kpe_cache, return_lse=True)

    for b in range(batch):
        L = kv_lens[b]
        rows = []
        for pi_ in range(pages_per[b]):
            pg = int(kv_indices[int(kv_indptr[b]) + pi_])
            n = min(page, L - pi_ * page)
            rows.append((ckv_cache[pg, :n], kpe_cache[pg, :n]))
        ckv = torch.cat([r[0] for r in rows])
        kpe = torch.cat([r[1] for r in rows])
        s, e = int(qo_indptr[b]), int(qo_indptr[b + 1])
        ref = mla_ref(q_nope[s:e], q_pe[s:e], ckv, kpe, sm_scale, causal,
                      e - s, L)
        torch.testing.assert_close(out[s:e].float(), ref, atol=3e-2, rtol=3e-2,
                                   msg=f"req {b}")
    assert lse.isfinite().all()
