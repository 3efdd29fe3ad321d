"""fp8-KV MFMA decode perf probe: BASELINE decode config (bs=16, kv=1024,
GQA-8, hd128) with bf16 vs fp8 e4m3 KV cache — fp8 halves KV HBM bytes."""
import torch, time
import flashinfer_amd as fi


def run(kv_dtype, B=16, L=1024, force=None, Hq=64, Hkv=8):
    import flashinfer_amd.decode as dec
    sv = (dec._FUSED_MAX_KV, dec._MFMA_MAX_KV, dec._MFMA_MAX_KV_F8)
    if force == "vector":
        dec._FUSED_MAX_KV = dec._MFMA_MAX_KV = dec._MFMA_MAX_KV_F8 = 0
    elif force == "fused":
        dec._MFMA_MAX_KV = dec._MFMA_MAX_KV_F8 = 0
    torch.manual_seed(0)
    D, page = 128, 16
    npages = B * (L // page)
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    if kv_dtype != torch.bfloat16:
        kc = kc.clamp(-8, 8).to(kv_dtype)
        vc = vc.clamp(-8, 8).to(kv_dtype)
    indptr = torch.arange(0, B + 1, dtype=torch.int32, device="cuda") * (L // page)
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((B,), page, dtype=torch.int32, device="cuda")
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(128 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    try:
        w.plan(indptr, indices, lpl, Hq, Hkv, D, page,
               q_data_type=torch.bfloat16, kv_data_type=kv_dtype)
    finally:
        dec._FUSED_MAX_KV, dec._MFMA_MAX_KV, dec._MFMA_MAX_KV_F8 = sv
    if force == "fused":
        assert w._fused and not w._fused_mfma
    elif force is None:
        assert w._fused_mfma, f"route not mfma for {kv_dtype}"
    for _ in range(20):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    iters = 300
    t0 = time.perf_counter()
    for _ in range(iters):
        out = w.run(q, (kc, vc))
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    kv_bytes = 2 * npages * page * Hkv * D * kc.element_size()
    print(f"bs={B} kv={L} G={Hq // Hkv} {str(kv_dtype):24s} {us:7.2f} us  "
          f"{kv_bytes / (us * 1e-6) / 1e12:5.2f} TB/s KV  "
          f"route={'mfma' if w._fused_mfma else ('fused' if w._fused else 'tc/vec')}")
    return out


for B, L in [(16, 1024), (64, 1024), (128, 4096)]:
    run(torch.bfloat16, B, L)
for B, L in [(16, 1024), (128, 4096), (32, 16384), (8, 32768)]:
    run(torch.float8_e4m3fn, B, L)
# small groups: GQA-6 bf16, GROUP-4 fp8 short (mfma vs fused A/B)
run(torch.bfloat16, 16, 4096, Hq=48, Hkv=8)
run(torch.bfloat16, 16, 4096, Hq=48, Hkv=8, force="vector")
run(torch.float8_e4m3fn, 16, 1024, Hq=32, Hkv=8)
run(torch.float8_e4m3fn, 16, 1024, Hq=32, Hkv=8, force="fused")
run(torch.float8_e4m3fn, 16, 1024, Hq=32, Hkv=32)  # MHA fp8
