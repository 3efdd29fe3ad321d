"""GPU numerics: paged-cache append + state merge kernels."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("kv_layout", ["NHD", "HND"])
@pytest.mark.parametrize("page_size", [1, 16, 17])
def test_append_paged_kv_cache(kv_layout, page_size):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    H, D = 4, 128
    seq_lens_new = [45, 8, 25, 22]
    past = [10, 0, 3, 1]
    total = [p + n for p, n in zip(past, seq_lens_new)]
    batch = len(total)
    pages_per = [(t + page_size - 1) // page_size for t in total]
    indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)),
                          dtype=torch.int32, device="cuda")
    npages = int(indptr[-1])
    perm = torch.randperm(npages, dtype=torch.int32, device="cuda")
    indices = perm
    last_page_len = torch.tensor(
        [(t - 1) % page_size + 1 for t in total], dtype=torch.int32, device="cuda"
    )
    if kv_layout == "NHD":
        shape = (npages, page_size, H, D)
    else:
        shape = (npages, H, page_size, D)
    k_cache = torch.zeros(shape, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.zeros(shape, dtype=torch.bfloat16, device="cuda")

    nnz = sum(seq_lens_new)
    append_indptr = torch.tensor(
        [0] + list(torch.cumsum(torch.tensor(seq_lens_new), 0)),
        dtype=torch.int32, device="cuda",
    )
    seq_lens_t = torch.tensor(total, dtype=torch.int32, device="cuda")
    k_new = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    v_new = torch.randn(nnz, H, D, dtype=torch.bfloat16, device="cuda")
    bi, pos = fi.get_batch_indices_positions(append_indptr, seq_lens_t, nnz)
    fi.append_paged_kv_cache(
        k_new, v_new, bi, pos, (k_cache, v_cache), indices, indptr,
        last_page_len, kv_layout,
    )
    # verify a scattering of tokens
    for b in range(batch):
        for i_local in [0, seq_lens_new[b] - 1]:
            i_global = int(append_indptr[b]) + i_local
            p = past[b] + i_local
            page = int(indices[int(indptr[b]) + p // page_size])
            entry = p % page_size
            if kv_layout == "NHD":
                got = k_cache[page, entry, :, :]
            else:
                got = k_cache[page, :, entry, :]
            torch.testing.assert_close(got, k_new[i_global])


def test_merge_state_matches_reference():
    import flashinfer_amd as fi

    torch.manual_seed(0)
    n, h, d = 33, 8, 128
    v_a = torch.randn(n, h, d, device="cuda", dtype=torch.float32)
    v_b = torch.randn(n, h, d, device="cuda", dtype=torch.float32)
    s_a = torch.randn(n, h, device="cuda") * 4
    s_b = torch.randn(n, h, device="cuda") * 4
    v, s = fi.merge_state(v_a, s_a, v_b, s_b)
    # reference (s in log2 scale, see reference trace/templates/cascade.py)
    import math

    s_an, s_bn = s_a * math.log(2), s_b * math.log(2)
    m = torch.maximum(s_an, s_bn)
    wa, wb = torch.exp(s_an - m), torch.exp(s_bn - m)
    ref_v = (v_a * wa[..., None] + v_b * wb[..., None]) / (wa + wb)[..., None]
    ref_s = (m + torch.log(wa + wb)) / math.log(2)
    torch.testing.assert_close(v, ref_v, atol=1e-3, rtol=1e-3)
    torch.testing.assert_close(s, ref_s, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("ns", [1, 2, 9])
def test_merge_states_uniform(ns):
    import flashinfer_amd as fi
    import math

    torch.manual_seed(0)
    n, h, d = 17, 4, 64
    v = torch.randn(n, ns, h, d, device="cuda", dtype=torch.float32)
    s = torch.randn(n, ns, h, device="cuda") * 3
    v_m, s_m = fi.merge_states(v, s)
    s_nat = s * math.log(2)
    m = s_nat.max(dim=1, keepdim=True).values
    w = torch.exp(s_nat - m)
    ref = (v * w[..., None]).sum(1) / w.sum(1)[..., None]
    torch.testing.assert_close(v_m, ref, atol=1e-3, rtol=1e-3)
    ref_s = (m.squeeze(1) + torch.log(w.sum(1))) / math.log(2)
    torch.testing.assert_close(s_m, ref_s, atol=1e-3, rtol=1e-3)


def test_merge_state_in_place():
    """In-place LSE merge (cascade primitive) vs the functional merge_state,
    including the optional per-position mask."""
    import flashinfer_amd as fi

    torch.manual_seed(6)
    n, h, d = 16, 8, 64
    va = torch.randn(n, h, d, dtype=torch.bfloat16, device="cuda")
    sa = torch.randn(n, h, device="cuda")
    vb = torch.randn(n, h, d, dtype=torch.bfloat16, device="cuda")
    sb = torch.randn(n, h, device="cuda")
    ref_v, ref_s = fi.merge_state(va.clone(), sa.clone(), vb, sb)
    v, s = va.clone(), sa.clone()
    fi.merge_state_in_place(v, s, vb, sb)
    torch.testing.assert_close(v.float(), ref_v.float(), atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(s, ref_s, atol=1e-4, rtol=1e-4)
    # masked positions keep their original state
    mask = torch.zeros(n, dtype=torch.bool, device="cuda")
    mask[: n // 2] = True
    v2, s2 = va.clone(), sa.clone()
    fi.merge_state_in_place(v2, s2, vb, sb, mask=mask)
    torch.testing.assert_close(v2[: n // 2].float(), ref_v[: n // 2].float(),
                               atol=2e-2, rtol=2e-2)
    assert torch.equal(v2[n // 2:], va[n // 2:])
    assert torch.equal(s2[n // 2:], sa[n // 2:])
