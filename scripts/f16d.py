import sys
from pathlib import Path
sys.path.insert(0, "/root/repo")
import time
import torch
import flashinfer_amd as fi

def bench(dtype, bs=256, kv=32768, Hq=32, Hkv=8, D=128, page=16):
    pages_per = kv // page
    npages = bs * pages_per
    indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
    indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=dtype, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=dtype, device="cuda")
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, last, Hq, Hkv, D, page, q_data_type=dtype)
    out = torch.empty_like(q)
    fn = lambda: w.run(q, (kc, vc), out=out)
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    tb = bs * kv * Hkv * D * 2 * 2 / dt / 1e12
    print(f"decode {dtype} bs={bs} kv={kv}: {dt*1e6:.0f} us  {tb:.2f} TB/s")

bench(torch.float16)
bench(torch.bfloat16)
