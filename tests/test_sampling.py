"""GPU tests: sampling suite — exact checks where deterministic, set-membership
and statistical checks for the samplers."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_softmax_matches_torch():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    logits = torch.randn(13, 32000, device="cuda") * 4
    out = sampling.softmax(logits, temperature=0.7)
    ref = torch.softmax(logits / 0.7, dim=-1)
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-4)


def test_sampling_from_probs_distribution():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    g = torch.Generator(device="cuda").manual_seed(42)
    probs = torch.tensor([[0.5, 0.25, 0.25, 0.0]], device="cuda").repeat(4096, 1)
    ids = sampling.sampling_from_probs(probs, generator=g)
    counts = torch.bincount(ids.long(), minlength=4).float() / 4096
    assert abs(counts[0] - 0.5) < 0.05
    assert counts[3] == 0
    # argmax case: one-hot
    probs2 = torch.zeros(8, 100, device="cuda")
    probs2[:, 7] = 1.0
    assert (sampling.sampling_from_probs(probs2, generator=g) == 7).all()


def test_sampling_from_logits():
    from flashinfer_amd import sampling

    g = torch.Generator(device="cuda").manual_seed(0)
    logits = torch.full((16, 1000), -100.0, device="cuda")
    logits[:, 33] = 10.0
    assert (sampling.sampling_from_logits(logits, generator=g) == 33).all()


@pytest.mark.parametrize("k", [1, 5, 100])
def test_top_k_sampling_membership(k):
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    g = torch.Generator(device="cuda").manual_seed(1)
    probs = torch.softmax(torch.randn(64, 2048, device="cuda") * 2, -1)
    for _ in range(5):
        ids = sampling.top_k_sampling_from_probs(probs, k, generator=g)
        topk = probs.topk(k, dim=-1).indices
        ok = (ids[:, None] == topk).any(-1)
        assert ok.all(), f"sampled outside top-{k}"


def test_top_p_sampling_membership():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    g = torch.Generator(device="cuda").manual_seed(2)
    p = 0.6
    probs = torch.softmax(torch.randn(64, 512, device="cuda") * 3, -1)
    sorted_p, idx = probs.sort(-1, descending=True)
    cum = sorted_p.cumsum(-1)
    # membership set: tokens whose "mass strictly above them" < p
    above = cum - sorted_p
    in_set = torch.zeros_like(probs, dtype=torch.bool)
    in_set.scatter_(1, idx, above < p)
    for _ in range(5):
        ids = sampling.top_p_sampling_from_probs(probs, p, generator=g)
        assert in_set.gather(1, ids.long()[:, None]).all()


def test_min_p_sampling_membership():
    from flashinfer_amd import sampling

    g = torch.Generator(device="cuda").manual_seed(3)
    probs = torch.softmax(torch.randn(32, 256, device="cuda") * 3, -1)
    mp = 0.2
    ids = sampling.min_p_sampling_from_probs(probs, mp, generator=g)
    thresh = probs.max(-1).values * mp
    assert (probs.gather(1, ids.long()[:, None]).squeeze(1) >= thresh * 0.999).all()


def test_top_k_renorm_matches_sort_reference():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    probs = torch.softmax(torch.randn(16, 1024, device="cuda") * 2, -1)
    k = 10
    out = sampling.top_k_renorm_probs(probs, k)
    kth = probs.topk(k, -1).values[:, -1:]
    ref = torch.where(probs >= kth, probs, torch.zeros_like(probs))
    ref = ref / ref.sum(-1, keepdim=True)
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-4)


def test_top_k_mask_logits():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    logits = torch.randn(8, 512, device="cuda") * 5
    out = sampling.top_k_mask_logits(logits, 20)
    kth = logits.topk(20, -1).values[:, -1:]
    assert ((out == float("-inf")) == (logits < kth)).all()
    torch.testing.assert_close(out[out != float("-inf")], logits[logits >= kth])


def test_top_p_renorm():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    probs = torch.softmax(torch.randn(8, 256, device="cuda") * 3, -1)
    p = 0.5
    out = sampling.top_p_renorm_probs(probs, p)
    # kept set must be the minimal top mass >= p (ties aside): verify each
    # row: sum of kept original probs >= p, and removing the smallest kept
    # element drops below p
    kept = out > 0
    mass = (probs * kept).sum(-1)
    assert (mass >= p - 1e-5).all()
    torch.testing.assert_close(out.sum(-1), torch.ones(8, device="cuda"),
                               atol=1e-5, rtol=1e-5)


def test_indices_indirection():
    from flashinfer_amd import sampling

    g = torch.Generator(device="cuda").manual_seed(4)
    probs = torch.zeros(2, 64, device="cuda")
    probs[0, 5] = 1.0
    probs[1, 9] = 1.0
    indices = torch.tensor([1, 1, 0], dtype=torch.int32, device="cuda")
    ids = sampling.sampling_from_probs(probs, indices=indices, generator=g)
    assert ids.tolist() == [9, 9, 5]


def test_chain_speculative_sampling():
    from flashinfer_amd import sampling

    torch.manual_seed(0)
    g = torch.Generator(device="cuda").manual_seed(5)
    B, n, V = 4, 3, 128
    # draft == target -> always accept, final token sampled from target[n]
    probs = torch.softmax(torch.randn(B, n, V, device="cuda"), -1)
    draft_ids = probs.argmax(-1).int()
    target = torch.cat([probs, torch.softmax(torch.randn(B, 1, V, device="cuda"), -1)],
                       dim=1)
    out, acc, emit = sampling.chain_speculative_sampling(
        probs, draft_ids, target, generator=g
    )
    assert (out[:, :n] == draft_ids).all()
    assert (acc == n).all() and (emit == n).all()
    assert (out[:, n] >= 0).all()
    # draft puts mass on a token the target gives 0 -> reject at step 0
    dp = torch.zeros(1, 1, V, device="cuda")
    dp[0, 0, 3] = 1.0
    tp = torch.zeros(1, 2, V, device="cuda")
    tp[0, 0, 7] = 1.0
    tp[0, 1, 11] = 1.0
    out, acc, emit = sampling.chain_speculative_sampling(
        dp, torch.tensor([[3]], dtype=torch.int32, device="cuda"), tp, generator=g
    )
    assert out[0, 0] == 7 and out[0, 1] == -1
    assert acc[0] == 0


def test_renorm_large_vocab_exact():
    """128k vocab exercises all three radix-histogram levels; the threshold
    must match torch.topk exactly (bit-exact tau)."""
    from flashinfer_amd import sampling

    torch.manual_seed(11)
    B, V = 8, 128256
    logits = torch.randn(B, V, device="cuda") * 6
    probs = torch.softmax(logits, -1)
    for k in (1, 50, 1000):
        out = sampling.top_k_renorm_probs(probs, k)
        kth = probs.topk(k, -1).values[:, -1:]
        ref = torch.where(probs >= kth, probs, torch.zeros_like(probs))
        ref = ref / ref.sum(-1, keepdim=True)
        torch.testing.assert_close(out, ref, atol=1e-6, rtol=1e-5)
        mask = sampling.top_k_mask_logits(logits, k)
        kth_l = logits.topk(k, -1).values[:, -1:]
        assert ((mask == float("-inf")) == (logits < kth_l)).all(), k
    # per-row k tensor
    ks = torch.tensor([1, 7, 50, 333, 1000, 5, 2, 64], dtype=torch.int32,
                      device="cuda")
    out = sampling.top_k_renorm_probs(probs, ks)
    for b in range(B):
        kth = probs[b].topk(int(ks[b]), -1).values[-1]
        ref = torch.where(probs[b] >= kth, probs[b],
                          torch.zeros_like(probs[b]))
        torch.testing.assert_close(out[b], ref / ref.sum(), atol=1e-6,
                                   rtol=1e-5)


def test_top_p_renorm_large_vocab():
    from flashinfer_amd import sampling

    torch.manual_seed(12)
    B, V = 8, 128256
    probs = torch.softmax(torch.randn(B, V, device="cuda") * 4, -1)
    pth = 0.9
    out = sampling.top_p_renorm_probs(probs, pth)
    # kept set = minimal top mass >= pth: verify via sort
    sp, si = probs.sort(-1, descending=True)
    cum = sp.cumsum(-1)
    nkeep = (cum < pth).sum(-1) + 1
    for b in range(B):
        kept = out[b] > 0
        assert kept.sum() == nkeep[b], (b, int(kept.sum()), int(nkeep[b]))
    torch.testing.assert_close(out.sum(-1), torch.ones(B, device="cuda"),
                               atol=1e-5, rtol=1e-5)
