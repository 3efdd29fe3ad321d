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
