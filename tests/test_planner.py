"""CPU tests for host-side planning logic (no GPU required)."""
import math

import pytest
import torch

from flashinfer_amd.decode import _plan_chunks
from flashinfer_amd.utils import WorkspaceAllocator
from flashinfer_amd.page import get_seq_lens


@pytest.mark.parametrize("batch", [1, 7, 256])
@pytest.mark.parametrize("max_len", [1, 300, 32768])
@pytest.mark.parametrize("num_kv_heads", [1, 8])
def test_plan_chunks_covers_everything(batch, max_len, num_kv_heads):
    torch.manual_seed(0)
    kv_lens = [int(x) for x in torch.randint(1, max_len + 1, (batch,))]
    page_size = 16
    chunk, work_req, work_chunk, merge_indptr = _plan_chunks(
        kv_lens, num_kv_heads, page_size
    )
    assert chunk % page_size == 0
    assert len(work_req) == len(work_chunk) == merge_indptr[-1]
    assert len(merge_indptr) == batch + 1
    # each request covered by exactly ceil(len/chunk) chunks, ids 0..n-1
    for b, L in enumerate(kv_lens):
        items = [
            work_chunk[i]
            for i in range(merge_indptr[b], merge_indptr[b + 1])
        ]
        assert all(work_req[i] == b for i in range(merge_indptr[b], merge_indptr[b + 1]))
        n = max(1, math.ceil(L / chunk))
        assert items == list(range(n))
        # chunks cover [0, L): last chunk start < L
        assert (n - 1) * chunk < L or L == 0


def test_workspace_allocator_alignment_and_overflow():
    buf = torch.zeros(1024, dtype=torch.uint8)
    a = WorkspaceAllocator(buf)
    t1 = a.alloc(10, torch.uint8, (10,))
    t2 = a.alloc(16, torch.float32, (4,))
    assert t2.data_ptr() % 256 == 0 or t2.data_ptr() == buf.data_ptr() + 256
    with pytest.raises(RuntimeError):
        a.alloc(4096, torch.uint8, (4096,))


def test_get_seq_lens():
    indptr = torch.tensor([0, 2, 5, 5])
    last = torch.tensor([3, 16, 9])
    # page_size 16: req0 = 1*16+3 = 19; req1 = 2*16+16 = 48; req2 empty = 0
    lens = get_seq_lens(indptr, last, 16)
    assert lens.tolist() == [19, 48, 0]


def test_fastdiv_magic():
    from flashinfer_amd import _lib

    if not _lib.has_ext():
        pytest.skip("native extension not built")
    ext = _lib.get_ext()
    vals = [0, 1, 2, 3, 15, 16, 17, 255, 256, 1000, 12345, 2**31 - 1, 2**32 - 1]
    for d in [1, 2, 3, 5, 7, 15, 16, 17, 64, 100, 128, 255, 257, 32768, 65535]:
        got = ext.debug_fastdiv(d, vals)
        assert got == [v // d for v in vals], f"divisor {d}"


# ---- property-based invariants (hypothesis) ----
try:
    from hypothesis import given, settings, strategies as st

    HAVE_HYP = True
except ImportError:  # pragma: no cover
    HAVE_HYP = False


if HAVE_HYP:
    @given(
        kv_lens=st.lists(st.integers(min_value=0, max_value=100000),
                         min_size=1, max_size=64),
        num_kv_heads=st.sampled_from([1, 2, 4, 8, 16]),
        page_size=st.sampled_from([1, 16, 32, 64]),
    )
    @settings(max_examples=200, deadline=None)
    def test_plan_chunks_invariants(kv_lens, num_kv_heads, page_size):
        from flashinfer_amd.decode import _plan_chunks

        chunk, work_req, work_chunk, merge_indptr = _plan_chunks(
            kv_lens, num_kv_heads, page_size)
        assert chunk % page_size == 0 and chunk >= page_size
        assert len(work_req) == len(work_chunk) == merge_indptr[-1]
        assert merge_indptr[0] == 0
        # every request gets at least one item; chunks tile the kv length
        for b, L in enumerate(kv_lens):
            items = [c for r, c in zip(work_req, work_chunk) if r == b]
            assert items == list(range(len(items)))
            assert len(items) == max(1, -(-L // chunk))
            assert merge_indptr[b + 1] - merge_indptr[b] == len(items)

    @given(
        qo_lens=st.lists(st.integers(min_value=0, max_value=5000),
                         min_size=1, max_size=32),
        group=st.sampled_from([1, 2, 4, 8]),
        causal=st.booleans(),
    )
    @settings(max_examples=200, deadline=None)
    def test_plan_tiles_invariants(qo_lens, group, causal):
        from flashinfer_amd.prefill import _plan_tiles

        cta_q, tile_req, tile_qstart = _plan_tiles(qo_lens, group, causal)
        assert cta_q in (128, 256)
        assert len(tile_req) == len(tile_qstart)
        # tiles exactly cover every request's packed rows
        from collections import defaultdict

        starts = defaultdict(list)
        for r, s in zip(tile_req, tile_qstart):
            starts[r].append(s)
        for b, L in enumerate(qo_lens):
            pk = max(L * group, 1)
            expect = list(range(0, pk, cta_q))
            assert sorted(starts[b]) == expect

    @given(n=st.integers(min_value=1, max_value=2**31 - 1),
           d=st.integers(min_value=1, max_value=2**31 - 1))
    @settings(max_examples=300, deadline=None)
    def test_fastdiv_magic_matches_python(n, d):
        import flashinfer_amd as fi
        from flashinfer_amd._lib import has_ext, get_ext

        if not has_ext():
            return
        res = get_ext().debug_fastdiv(d, [n])
        assert res[0] == n // d
