"""GPU tests: cascade (shared-prefix) attention and block-sparse attention
against dense fp32 references."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_cascade_two_level_matches_dense():
    import flashinfer_amd as fi
    from flashinfer_amd.cascade import MultiLevelCascadeAttentionWrapper

    torch.manual_seed(0)
    Hq, Hkv, D, page = 8, 2, 128, 16
    batch = 4
    shared_len, unique_len, qo_len = 128, 64, 32
    # unified page table: shared pages first, then per-request unique pages
    sp = shared_len // page
    up = unique_len // page
    npages = sp + batch * up
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    nnz = batch * qo_len
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")

    qo_indptr_top = torch.tensor([0, nnz], dtype=torch.int32, device="cuda")
    qo_indptr_bot = torch.arange(0, nnz + 1, qo_len, dtype=torch.int32, device="cuda")
    kv_indptr_top = torch.tensor([0, sp], dtype=torch.int32, device="cuda")
    kv_indices_top = torch.arange(sp, dtype=torch.int32, device="cuda")
    kv_indptr_bot = torch.arange(0, batch * up + 1, up, dtype=torch.int32,
                                 device="cuda")
    kv_indices_bot = sp + torch.arange(batch * up, dtype=torch.int32, device="cuda")
    lp_top = torch.tensor([page], dtype=torch.int32, device="cuda")
    lp_bot = torch.full((batch,), page, dtype=torch.int32, device="cuda")

    ws = torch.empty(32 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = MultiLevelCascadeAttentionWrapper(2, ws, "NHD")
    w.plan([qo_indptr_top, qo_indptr_bot], [kv_indptr_top, kv_indptr_bot],
           [kv_indices_top, kv_indices_bot], [lp_top, lp_bot],
           Hq, Hkv, D, page, causal=True)
    out = w.run(q, (k_cache, v_cache))

    # dense reference per request: kv = shared + unique, causal over suffix
    g = Hq // Hkv
    k_flat = k_cache.float().view(-1, Hkv, D)
    v_flat = v_cache.float().view(-1, Hkv, D)
    for b in range(batch):
        kv_rows = torch.cat([
            k_flat[:shared_len],
            k_flat[shared_len + b * unique_len: shared_len + (b + 1) * unique_len],
        ])
        v_rows = torch.cat([
            v_flat[:shared_len],
            v_flat[shared_len + b * unique_len: shared_len + (b + 1) * unique_len],
        ])
        L = shared_len + unique_len
        qb = q[b * qo_len:(b + 1) * qo_len].float()
        kf = kv_rows.repeat_interleave(g, dim=1)
        vf = v_rows.repeat_interleave(g, dim=1)
        logits = torch.einsum("mhd,lhd->hml", qb, kf) / math.sqrt(D)
        qpos = torch.arange(qo_len, device="cuda")[:, None]
        kpos = torch.arange(L, device="cuda")[None, :]
        # causal on the unique suffix: q i may see shared + unique[0..i+off]
        mask = kpos > qpos + (L - qo_len)
        logits = logits.masked_fill(mask[None], float("-inf"))
        p = torch.softmax(logits, -1)
        ref = torch.einsum("hml,lhd->mhd", p, vf)
        torch.testing.assert_close(out[b * qo_len:(b + 1) * qo_len].float(), ref,
                                   atol=3e-2, rtol=3e-2)


def test_block_sparse_matches_dense():
    import flashinfer_amd as fi
    from flashinfer_amd.sparse import BlockSparseAttentionWrapper

    torch.manual_seed(0)
    M, N, R, C = 128, 256, 16, 16
    Hq, Hkv, D = 4, 4, 64
    MB, NB = M // R, N // C
    dense_mask = torch.rand(MB, NB, device="cuda") > 0.5
    dense_mask[:, 0] = True  # every row attends something
    indptr = torch.zeros(MB + 1, dtype=torch.int32, device="cuda")
    indptr[1:] = dense_mask.sum(1).cumsum(0)
    indices = torch.nonzero(dense_mask)[:, 1].int()

    q = torch.randn(M, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")

    ws = torch.empty(32 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = BlockSparseAttentionWrapper(ws)
    w.plan(indptr, indices, M, N, R, C, Hq, Hkv, D)
    out = w.run(q, k, v)

    # dense reference with -inf outside the block pattern
    token_mask = dense_mask.repeat_interleave(R, 0).repeat_interleave(C, 1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), k.float()) / math.sqrt(D)
    logits = logits.masked_fill(~token_mask[None], float("-inf"))
    p = torch.softmax(logits, -1)
    ref = torch.einsum("hml,lhd->mhd", p, v.float())
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_variable_block_sparse_matches_dense():
    from flashinfer_amd.sparse import VariableBlockSparseAttentionWrapper

    torch.manual_seed(1)
    Hkv, G, D = 2, 2, 128
    Hq = Hkv * G
    MB, NB = 3, 4
    # per-head variable block sizes summing to the same seq lens
    row_sz = torch.tensor([[32, 64, 32], [16, 96, 16]], dtype=torch.int32)
    col_sz = torch.tensor([[64, 32, 16, 16], [32, 32, 32, 32]], dtype=torch.int32)
    qo_len, kv_len = 128, 128
    assert (row_sz.sum(1) == qo_len).all() and (col_sz.sum(1) == kv_len).all()
    mask_map = torch.rand(Hkv, MB, NB) > 0.4
    mask_map[:, :, 0] = True  # every row block attends something

    q = torch.randn(Hq, qo_len, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(Hkv, kv_len, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(Hkv, kv_len, D, dtype=torch.bfloat16, device="cuda")

    ws = torch.empty(64 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = VariableBlockSparseAttentionWrapper(ws)
    w.plan(mask_map, row_sz, col_sz, Hq, Hkv, D)
    out, lse = w.run(q, k, v, return_lse=True)
    assert out.shape == (Hq, qo_len, D) and lse.shape == (Hq, qo_len)

    # dense fp32 reference: expand the per-head block mask to token level
    for h in range(Hkv):
        tok_mask = mask_map[h].repeat_interleave(
            row_sz[h].long(), 0).repeat_interleave(col_sz[h].long(), 1).cuda()
        for gi in range(G):
            qh = q[h * G + gi].float()
            logits = qh @ k[h].float().t() / math.sqrt(D)
            logits = logits.masked_fill(~tok_mask, float("-inf"))
            ref = torch.softmax(logits, -1) @ v[h].float()
            torch.testing.assert_close(out[h * G + gi].float(), ref,
                                       atol=3e-2, rtol=3e-2)


def test_block_sparse_with_element_mask():
    from flashinfer_amd.sparse import BlockSparseAttentionWrapper

    torch.manual_seed(7)
    M, N, R, C = 64, 128, 16, 16
    Hq, Hkv, D = 2, 2, 64
    MB, NB = M // R, N // C
    dense_mask = torch.rand(MB, NB, device="cuda") > 0.4
    dense_mask[:, 0] = True
    indptr = torch.zeros(MB + 1, dtype=torch.int32, device="cuda")
    indptr[1:] = dense_mask.sum(1).cumsum(0)
    indices = torch.nonzero(dense_mask)[:, 1].int()
    nnz = int(indptr[-1])
    elem = torch.rand(nnz, R, C, device="cuda") > 0.3
    elem[:, :, 0] = True  # keep every row non-empty

    q = torch.randn(M, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(N, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(32 << 20, dtype=torch.uint8, device="cuda")
    w = BlockSparseAttentionWrapper(ws)
    w.plan(indptr, indices, M, N, R, C, Hq, Hkv, D, mask=elem)
    out = w.run(q, k, v)

    # dense reference: block mask AND element mask
    token_mask = torch.zeros(M, N, dtype=torch.bool, device="cuda")
    for br in range(MB):
        for j in range(int(indptr[br]), int(indptr[br + 1])):
            bc = int(indices[j])
            token_mask[br * R:(br + 1) * R, bc * C:(bc + 1) * C] = elem[j]
    logits = torch.einsum("mhd,lhd->hml", q.float(), k.float()) / math.sqrt(D)
    logits = logits.masked_fill(~token_mask[None], float("-inf"))
    p = torch.softmax(logits, -1)
    ref = torch.einsum("hml,lhd->mhd", p, v.float())
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
