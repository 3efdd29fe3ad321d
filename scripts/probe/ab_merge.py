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


import flashinfer_amd.decode as dec

def run_route(B, L, Hq, Hkv, route):
    sv = (dec._FUSED_MAX_KV, dec._MFMA_MAX_KV)
    if route == "mfma":
        dec._FUSED_MAX_KV = 0
    torch.manual_seed(0)
    D, page = 128, 16
    npages = B * max(1, L // page)
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    indptr = torch.arange(0, B + 1, dtype=torch.int32, device="cuda") * (L // page)
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((B,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    try:
        w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
               q_data_type=torch.bfloat16)
    finally:
        dec._FUSED_MAX_KV, dec._MFMA_MAX_KV = sv
    if route == "mfma":
        assert w._fused_mfma
    for _ in range(30):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(500):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 500 * 1e6
    rt = ("mfma" if w._fused_mfma else
          ("fused" if w._fused else ("tc" if w._tc else "vec")))
    print(f"bs={B} kv={L} G={Hq // Hkv}: route={rt:6s} {us:7.2f} us")


for B, L in [(16, 1024), (32, 512), (64, 2048), (8, 256)]:
    run_route(B, L, 32, 8, "auto")
    run_route(B, L, 32, 8, "mfma")
for B, L in [(16, 1024), (32, 512)]:
    for Hq in (16, 8):
        run_route(B, L, Hq, 8, "auto")
        run_route(B, L, Hq, 8, "mfma")
run_route(48, 1024, 32, 8, "auto")   # 384 units: fused vs
run_route(48, 1024, 32, 8, "mfma")


# fixed-vs-per-tile decomposition: bs=16 GQA-8, split fixed at 2
for L in (256, 512, 1024, 2048, 4096):
    run_route(16, L, 64, 8, "auto")
