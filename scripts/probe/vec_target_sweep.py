"""Sweep the split-vector decode items target at the bench decode config
(bs=256, kv=32768, GQA 32q/8kv) — the headline bench.py decode phase."""
import time

import torch

import flashinfer_amd as fi
import flashinfer_amd.decode as dec


def run(target):
    dec._TARGET_BLOCKS = target
    torch.manual_seed(0)
    B, Hq, Hkv, D, page, L = 256, 32, 8, 128, 16, 32768
    npages = B * (L // page)
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    indptr = torch.arange(0, B + 1, dtype=torch.int32, device="cuda") * (L // page)
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((B,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    assert not w._fused
    for _ in range(5):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    iters = 30
    t0 = time.perf_counter()
    for _ in range(iters):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    kv_bytes = 2 * npages * page * Hkv * D * 2
    print(f"target={target:6d}: {us:8.1f} us  {kv_bytes / (us * 1e-6) / 1e12:5.2f} TB/s")


for t in (4096, 8192, 16384, 32768, 65536):
    run(t)
run(8192)
