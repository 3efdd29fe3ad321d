import sys
from pathlib import Path
sys.path.insert(0, "/root/repo")
import time
import torch
import flashinfer_amd as fi

def bench_fp8(bs=256, kv=8192, Hq=32, Hkv=8, D=128, page=16):
    pages_per = (kv + page - 1) // page
    npages = bs * pages_per
    indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
    indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = (torch.randn(npages, page, Hkv, D, device="cuda") / 8).to(torch.float8_e4m3fn)
    vc = (torch.randn(npages, page, Hkv, D, device="cuda") / 8).to(torch.float8_e4m3fn)
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, last, Hq, Hkv, D, page, q_data_type=torch.bfloat16,
           kv_data_type=torch.float8_e4m3fn)
    for _ in range(5): w.run(q, (kc, vc))
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(20): w.run(q, (kc, vc))
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 20
    tb = bs * kv * Hkv * D * 2 / dt / 1e12
    print(f"fp8-KV decode bs={bs} kv={kv} G{Hq//Hkv}: {dt*1e6:.1f} us  {tb:.2f} TB/s")

bench_fp8()
