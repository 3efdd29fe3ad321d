import time
import torch
import flashinfer_amd as fi


def run(B, L, dt, inkernel, Hq=64, Hkv=8, force_split=None):
    torch.manual_seed(0)
    D, page = 128, 16
    npages = B * (L // page)
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    if dt != torch.bfloat16:
        kc = kc.clamp(-8, 8).to(dt); vc = vc.clamp(-8, 8).to(dt)
    indptr = torch.arange(0, B + 1, dtype=torch.int32, device="cuda") * (L // page)
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((B,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16, kv_data_type=dt)
    assert w._fused_mfma and w._mfma_split > 1
    if force_split is not None and force_split != w._mfma_split:
        from flashinfer_amd.utils import WorkspaceAllocator
        w._mfma_split = force_split
        alloc = WorkspaceAllocator(w._float_workspace_buffer)
        ni = B * force_split
        Hqh = q_heads = Hq
        w._mfma_tmp_v = alloc.alloc(ni * Hqh * D * 4, torch.float32,
                                    (ni, Hqh, D))
        w._mfma_tmp_s = alloc.alloc(ni * Hqh * 4, torch.float32, (ni, Hqh))
        w._mfma_merge_indptr = torch.arange(
            0, (B + 1) * force_split, force_split, dtype=torch.int32,
            device="cuda")
        w._mfma_counters = torch.zeros(B * Hkv, dtype=torch.int32,
                                       device="cuda")
    if not inkernel:
        w._mfma_counters = None
    for _ in range(30):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(500):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 500 * 1e6
    print(f"bs={B} kv={L} {str(dt)[6:]:16s} split={w._mfma_split} "
          f"{'inkernel' if inkernel else 'mergekrn'}: {us:7.2f} us")
    return out


o4 = run(16, 1024, torch.bfloat16, True)
for sp in (2, 8):
    o = run(16, 1024, torch.bfloat16, True, force_split=sp)
    torch.testing.assert_close(o, o4, atol=3e-2, rtol=3e-2)
    o = run(16, 1024, torch.bfloat16, False, force_split=sp)
    torch.testing.assert_close(o, o4, atol=3e-2, rtol=3e-2)
run(16, 1024, torch.bfloat16, True)
