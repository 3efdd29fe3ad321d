"""GPU: Minimax sparse attention pipeline vs dense fp32 references."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

BLOCK = 128


def test_msa_proxy_score_matches_dense():
    from flashinfer_amd.msa_ops import msa_proxy_score

    torch.manual_seed(0)
    Hq, Hkv, D = 4, 2, 128
    lens_q, lens_k = [64, 32], [300, 520]
    cq = torch.tensor([0, 64, 96], dtype=torch.int32, device="cuda")
    ck = torch.tensor([0, 300, 820], dtype=torch.int32, device="cuda")
    q = torch.randn(96, Hq, D, device="cuda").bfloat16()
    k = torch.randn(820, Hkv, D, device="cuda").bfloat16()
    out = msa_proxy_score(q, k, cq, ck, causal=True)
    T = max((L + BLOCK - 1) // BLOCK for L in lens_k)
    assert out.shape == (Hq, T, 96)
    G = Hq // Hkv
    # dense reference
    for b, (Lq, Lk, q0, k0) in enumerate(zip(lens_q, lens_k, [0, 64], [0, 300])):
        qb = q[q0:q0 + Lq].float()
        kb = k[k0:k0 + Lk].float()
        s = torch.einsum("qhd,khd->hqk", qb, kb.repeat_interleave(G, 1))
        qpos = torch.arange(Lq, device="cuda") + (Lk - Lq)
        kpos = torch.arange(Lk, device="cuda")
        s = s.masked_fill(kpos[None, None] > qpos[None, :, None], float("-inf"))
        nb = (Lk + BLOCK - 1) // BLOCK
        pad = nb * BLOCK - Lk
        sp = torch.nn.functional.pad(s, (0, pad), value=float("-inf"))
        ref = sp.view(Hq, Lq, nb, BLOCK).amax(-1)         # [Hq, Lq, nb]
        got = out[:, :nb, q0:q0 + Lq].permute(0, 2, 1)
        torch.testing.assert_close(got, ref, atol=2e-1, rtol=2e-2)
        assert (out[:, nb:, q0:q0 + Lq] == float("-inf")).all()
    # reduce_heads collapses the head axis with amax
    red = msa_proxy_score(q, k, cq, ck, causal=True, reduce_heads=True)
    torch.testing.assert_close(red, out.amax(0, keepdim=True))


def test_msa_topk_select():
    from flashinfer_amd.msa_ops import msa_topk_select

    torch.manual_seed(1)
    H, T, Q, K = 2, 24, 8, 6
    score = torch.randn(H, T, Q, device="cuda")
    valid = torch.randint(4, T + 1, (Q,), dtype=torch.int32, device="cuda")
    idx = msa_topk_select(score, K, num_valid_pages=valid,
                          force_begin_blocks=1, force_end_blocks=1)
    assert idx.shape == (Q, H, K) and idx.dtype == torch.int32
    for qi in range(Q):
        v = int(valid[qi])
        for h in range(H):
            sel = idx[qi, h]
            got = sel[sel >= 0].tolist()
            assert got == sorted(got)                    # ascending
            assert all(g < v for g in got)
            assert 0 in got and (v - 1) in got           # forced sink+local
            kef = min(K, v)
            assert len(got) == kef
            # the non-forced picks are the largest remaining scores
            rest = [g for g in got if g not in (0, v - 1)]
            s = score[h, :v, qi].clone()
            s[0] = s[v - 1] = float("inf")
            ref = torch.topk(s, kef).indices.tolist()
            assert set(got) == set(ref)


def test_msa_sparse_attention_full_blocks_match_dense():
    """With every block selected, sparse attention equals dense attention."""
    from flashinfer_amd.msa_ops import msa_sparse_attention

    torch.manual_seed(2)
    Hq, Hkv, D = 4, 2, 128
    Lq, Lk = 32, 512
    cq = torch.tensor([0, Lq], dtype=torch.int32, device="cuda")
    ck = torch.tensor([0, Lk], dtype=torch.int32, device="cuda")
    q = torch.randn(Lq, Hq, D, device="cuda").bfloat16()
    k = torch.randn(Lk, Hkv, D, device="cuda").bfloat16()
    v = torch.randn(Lk, Hkv, D, device="cuda").bfloat16()
    T = Lk // BLOCK
    q2k = torch.arange(T, dtype=torch.int32, device="cuda").expand(
        Hkv, Lq, T).contiguous()
    out, lse = msa_sparse_attention(q, k, v, q2k, cq, ck, causal=True,
                                    return_softmax_lse=True)
    G = Hq // Hkv
    kf = k.float().repeat_interleave(G, 1)
    vf = v.float().repeat_interleave(G, 1)
    logits = torch.einsum("qhd,khd->hqk", q.float(), kf) / math.sqrt(D)
    qpos = torch.arange(Lq, device="cuda") + (Lk - Lq)
    kpos = torch.arange(Lk, device="cuda")
    logits = logits.masked_fill(kpos[None, None] > qpos[None, :, None],
                                float("-inf"))
    p = torch.softmax(logits, -1)
    ref = torch.einsum("hqk,khd->qhd", p, vf)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    ref_lse = torch.logsumexp(logits, -1).t()
    torch.testing.assert_close(lse, ref_lse, atol=1e-2, rtol=1e-2)


def test_msa_sparse_attention_subset_blocks():
    """Selected-subset attention equals dense attention restricted to the
    selected tokens."""
    from flashinfer_amd.msa_ops import msa_sparse_attention

    torch.manual_seed(3)
    Hq, Hkv, D = 2, 2, 128
    Lq, Lk = 4, 640
    cq = torch.tensor([0, Lq], dtype=torch.int32, device="cuda")
    ck = torch.tensor([0, Lk], dtype=torch.int32, device="cuda")
    q = torch.randn(Lq, Hq, D, device="cuda").bfloat16()
    k = torch.randn(Lk, Hkv, D, device="cuda").bfloat16()
    v = torch.randn(Lk, Hkv, D, device="cuda").bfloat16()
    q2k = torch.tensor([0, 2, 4, -1], dtype=torch.int32, device="cuda").expand(
        Hkv, Lq, 4).contiguous()
    out = msa_sparse_attention(q, k, v, q2k, cq, ck, causal=False)
    sel = torch.cat([torch.arange(b * BLOCK, (b + 1) * BLOCK)
                     for b in [0, 2, 4]]).cuda()
    for h in range(Hq):
        logits = (q[:, h].float() @ k[sel, h].float().t()) / math.sqrt(D)
        ref = torch.softmax(logits, -1) @ v[sel, h].float()
        torch.testing.assert_close(out[:, h].float(), ref, atol=3e-2, rtol=3e-2)


def test_msa_sparse_decode():
    from flashinfer_amd.msa_ops import msa_sparse_decode_attention

    torch.manual_seed(4)
    Hq, Hkv, D, B = 4, 2, 128, 3
    Lk = 384
    ck = torch.arange(0, (B + 1) * Lk, Lk, dtype=torch.int32, device="cuda")
    q = torch.randn(B, Hq, D, device="cuda").bfloat16()
    k = torch.randn(B * Lk, Hkv, D, device="cuda").bfloat16()
    v = torch.randn(B * Lk, Hkv, D, device="cuda").bfloat16()
    T = Lk // BLOCK
    q2k = torch.arange(T, dtype=torch.int32, device="cuda").expand(
        Hkv, B, T).contiguous()
    out = msa_sparse_decode_attention(q, k, v, q2k, cu_seqlens_k=ck)
    G = Hq // Hkv
    for b in range(B):
        kb = k[b * Lk:(b + 1) * Lk].float().repeat_interleave(G, 1)
        vb = v[b * Lk:(b + 1) * Lk].float().repeat_interleave(G, 1)
        logits = torch.einsum("hd,khd->hk", q[b].float(), kb) / math.sqrt(D)
        ref = torch.einsum("hk,khd->hd", torch.softmax(logits, -1), vb)
        torch.testing.assert_close(out[b].float(), ref, atol=3e-2, rtol=3e-2)
