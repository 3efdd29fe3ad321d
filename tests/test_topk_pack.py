"""GPU tests: top_k + packbits."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("d,k", [(1000, 10), (32000, 256), (128, 128)])
def test_top_k_set_matches_torch(d, k):
    from flashinfer_amd.topk import top_k

    torch.manual_seed(0)
    x = torch.randn(8, d, device="cuda")
    v, i = top_k(x, k)
    ref_v, ref_i = torch.topk(x, k, dim=-1)
    # unordered: compare as sorted sets
    torch.testing.assert_close(v.sort(-1).values, ref_v.sort(-1).values)
    gathered = x.gather(1, i.long())
    torch.testing.assert_close(gathered, v)


def test_packbits():
    from flashinfer_amd.quantization import packbits

    torch.manual_seed(0)
    x = torch.rand(1000, device="cuda") > 0.5
    y = packbits(x)
    import numpy as np

    ref = np.packbits(x.cpu().numpy(), bitorder="little")
    assert (y.cpu().numpy() == ref).all()


def test_segment_packbits():
    from flashinfer_amd.quantization import segment_packbits

    torch.manual_seed(0)
    lens = [3, 0, 17, 64]
    indptr = torch.tensor([0, 3, 3, 20, 84], dtype=torch.int32, device="cuda")
    x = torch.rand(84, device="cuda") > 0.5
    y, y_indptr = segment_packbits(x, indptr)
    import numpy as np

    xs = x.cpu().numpy()
    off = 0
    for i, L in enumerate(lens):
        seg = xs[int(indptr[i]) : int(indptr[i + 1])]
        ref = np.packbits(seg, bitorder="little")
        got = y[int(y_indptr[i]) : int(y_indptr[i + 1])].cpu().numpy()
        assert (got == ref).all(), f"segment {i}"


def test_top_k_sorted_and_tiebreak():
    from flashinfer_amd.topk import top_k

    torch.manual_seed(1)
    x = torch.randn(4, 5000, device="cuda")
    v, i = top_k(x, 32, sorted=True)
    ref_v, _ = torch.topk(x, 32, dim=-1)
    torch.testing.assert_close(v, ref_v)
    assert (v[:, :-1] >= v[:, 1:]).all()
    # tie_break: constant input -> smallest / largest indices win
    xc = torch.zeros(2, 100, device="cuda")
    _, i1 = top_k(xc, 8, tie_break=1)
    assert set(i1[0].tolist()) == set(range(8))
    _, i2 = top_k(xc, 8, tie_break=2)
    assert set(i2[0].tolist()) == set(range(92, 100))


def test_top_k_ragged_transform():
    from flashinfer_amd.topk import top_k_ragged_transform

    torch.manual_seed(2)
    rows, d, k = 4, 300, 16
    x = torch.randn(rows, d, device="cuda")
    lengths = torch.tensor([300, 10, 128, 0], dtype=torch.int32, device="cuda")
    offsets = torch.tensor([0, 1000, 2000, 3000], dtype=torch.int32, device="cuda")
    out = top_k_ragged_transform(x, offsets, lengths, k)
    for r in range(rows):
        L = int(lengths[r])
        keff = min(k, L)
        got = out[r]
        assert (got[keff:] == -1).all()
        if keff:
            local = (got[:keff] - offsets[r]).long()
            ref = torch.topk(x[r, :L], keff).indices
            assert set(local.tolist()) == set(ref.tolist())


def test_top_k_page_table_transform():
    from flashinfer_amd.topk import top_k_page_table_transform

    torch.manual_seed(3)
    rows, d, k, page = 3, 64, 8, 4
    x = torch.randn(rows, d, device="cuda")
    lengths = torch.full((rows,), d, dtype=torch.int32, device="cuda")
    pt = torch.randperm(64, dtype=torch.int32, device="cuda").reshape(
        1, 64).repeat(rows, 1)
    out = top_k_page_table_transform(x, pt, lengths, k, page_size=page)
    for r in range(rows):
        ref_i = torch.topk(x[r], k).indices
        expect = {int(pt[r, i // page]) * page + int(i) % page
                  for i in ref_i.tolist()}
        assert set(out[r].tolist()) == expect


def test_top_k_varlen():
    from flashinfer_amd.topk import top_k_varlen

    torch.manual_seed(4)
    flat = torch.randn(500, device="cuda")
    offsets = torch.tensor([0, 200, 210, 500], dtype=torch.int32, device="cuda")
    v, i = top_k_varlen(flat, offsets=offsets, k=16)
    for r, (a, b) in enumerate(zip(offsets[:-1], offsets[1:])):
        seg = flat[a:b]
        kef = min(16, seg.numel())
        ref = torch.topk(seg, kef).values
        torch.testing.assert_close(v[r, :kef].sort(-1, descending=True).values,
                                   ref)
        assert (i[r, kef:] == -1).all()
