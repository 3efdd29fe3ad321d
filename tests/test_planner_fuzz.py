"""Fuzz the causal prefill tile planner (three-way: plain / short-tile
second CTAQ-128 launch / short-tile chaining): whatever path is chosen,
the emitted tiles must cover every packed q row of every request EXACTLY
once. Runs on CPU (plan() is host-side)."""
import random

import pytest
import torch


def _coverage(w, qo_lens, group, cta_q):
    """Collect (req, start, extent) from primary, chained and second-launch
    tile lists and assert exact coverage of [0, packed_len) per request."""
    rows = {b: torch.zeros(max(1, L * group), dtype=torch.int32)
            for b, L in enumerate(qo_lens)}

    def add(req_t, qst_t, extent):
        for b, st in zip(req_t.tolist(), qst_t.tolist()):
            if b < 0:
                continue
            pk = rows[b].numel()
            lo, hi = st, min(st + extent, pk)
            assert lo < hi, (b, st, extent, pk)
            rows[b][lo:hi] += 1

    add(w._tile_req, w._tile_qstart, cta_q)
    if getattr(w, "_tile_chain_req", None) is not None:
        add(w._tile_chain_req, w._tile_chain_qstart, cta_q)
    if getattr(w, "_tile_req2", None) is not None:
        add(w._tile_req2, w._tile_qstart2, 128)
    for b, cov in rows.items():
        assert (cov == 1).all(), (b, qo_lens[b],
                                  cov.min().item(), cov.max().item())


@pytest.mark.parametrize("seed", list(range(12)))
def test_causal_planner_tile_coverage(seed):
    import flashinfer_amd as fi

    rng = random.Random(seed)
    n = rng.randint(1, 24)
    group_choices = [(32, 8), (64, 8), (128, 128), (16, 8), (8, 8)]
    Hq, Hkv = rng.choice(group_choices)
    group = Hq // Hkv
    qo_lens = [rng.choice([1, 7, 63, 64, 128, 512, 1024, 2048, 4097])
               for _ in range(n)]
    kv_extra = [rng.choice([0, 0, 128, 1000]) for _ in range(n)]
    kv_lens = [q + e for q, e in zip(qo_lens, kv_extra)]
    page = 16
    qo_indptr = torch.zeros(n + 1, dtype=torch.int32)
    qo_indptr[1:] = torch.cumsum(torch.tensor(qo_lens), 0)
    kv_indptr = torch.zeros(n + 1, dtype=torch.int32)
    kv_indptr[1:] = torch.cumsum(
        torch.tensor([(k + page - 1) // page for k in kv_lens]), 0)
    lpl = torch.tensor([(k - 1) % page + 1 for k in kv_lens],
                       dtype=torch.int32)
    indices = torch.arange(int(kv_indptr[-1]), dtype=torch.int32)
    ws = torch.empty(1 << 20, dtype=torch.uint8)  # CPU workspace
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, indices, lpl, Hq, Hkv, 128, page,
           causal=True, q_data_type=torch.bfloat16)
    if getattr(w, "_split", False):
        return  # split path has its own chunk bookkeeping (tested on GPU)
    _coverage(w, qo_lens, group, w._plan_info["cta_q"])


def test_causal_planner_paths_reachable():
    """Both special paths trigger on their canonical shapes."""
    import flashinfer_amd as fi

    def plan(qo_lens, kv_lens, Hq, Hkv):
        page = 16
        n = len(qo_lens)
        qo_indptr = torch.zeros(n + 1, dtype=torch.int32)
        qo_indptr[1:] = torch.cumsum(torch.tensor(qo_lens), 0)
        kv_indptr = torch.zeros(n + 1, dtype=torch.int32)
        kv_indptr[1:] = torch.cumsum(
            torch.tensor([(k + page - 1) // page for k in kv_lens]), 0)
        lpl = torch.tensor([(k - 1) % page + 1 for k in kv_lens],
                          dtype=torch.int32)
        indices = torch.arange(int(kv_indptr[-1]), dtype=torch.int32)
        ws = torch.empty(64 << 20, dtype=torch.uint8)
        w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
        w.plan(qo_indptr, kv_indptr, indices, lpl, Hq, Hkv, 128, page,
               causal=True, q_data_type=torch.bfloat16)
        return w

    # bs=16 s=1024 GQA-4: shorts majority -> chaining
    w = plan([1024] * 16, [1024] * 16, 32, 8)
    assert getattr(w, "_tile_chain_req", None) is not None
    assert getattr(w, "_tile_req2", None) is None
    # bs=8 s=8192: shorts minority -> second CTAQ-128 launch
    w = plan([8192] * 8, [8192] * 8, 32, 8)
    if not getattr(w, "_split", False):
        assert getattr(w, "_tile_req2", None) is not None


@pytest.mark.parametrize("seed", list(range(8)))
def test_decode_chunk_planner_coverage(seed):
    """Split-vector decode planning: work items cover each request's chunks
    exactly, and merge_indptr allots n_chunks slots per request."""
    from flashinfer_amd.decode import _plan_chunks

    rng = random.Random(100 + seed)
    n = rng.randint(1, 40)
    kv_lens = [rng.choice([1, 15, 16, 17, 511, 4096, 32768, 100000])
               for _ in range(n)]
    page = rng.choice([1, 16, 32])
    chunk, work_req, work_chunk, merge_indptr = _plan_chunks(
        kv_lens, num_kv_heads=8, page_size=page)
    assert chunk % page == 0 and chunk > 0
    import math
    per_req = {}
    for b, c in zip(work_req, work_chunk):
        per_req.setdefault(b, []).append(c)
    for b, L in enumerate(kv_lens):
        nc = max(1, math.ceil(L / chunk))
        assert sorted(per_req[b]) == list(range(nc)), (b, L, chunk)
        assert merge_indptr[b + 1] - merge_indptr[b] == nc
    assert merge_indptr[0] == 0 and len(merge_indptr) == n + 1


def test_workspace_allocator():
    """Alignment, disjointness, overflow of the workspace carve-out used by
    every split-KV path's partial buffers."""
    from flashinfer_amd.utils import WorkspaceAllocator

    buf = torch.zeros(4096, dtype=torch.uint8)
    a = WorkspaceAllocator(buf)
    t1 = a.alloc(100, torch.uint8, (100,))
    t2 = a.alloc(256, torch.float32, (64,))
    t3 = a.alloc(16, torch.int32, (4,))
    # alignment: every allocation starts on a 256-byte boundary
    base = buf.data_ptr()
    for t in (t1, t2, t3):
        assert (t.data_ptr() - base) % 256 == 0
    # disjointness: writes don't alias
    t1.fill_(1)
    t2.fill_(2.0)
    t3.fill_(3)
    assert (t1 == 1).all() and (t2 == 2.0).all() and (t3 == 3).all()
    # overflow raises with a helpful message
    with pytest.raises(RuntimeError, match="workspace too small"):
        a.alloc(100000, torch.uint8, (100000,))
    # fresh allocator reuses from the start (wrappers re-plan this way)
    b = WorkspaceAllocator(buf)
    t4 = b.alloc(8, torch.uint8, (8,))
    assert t4.data_ptr() == base
