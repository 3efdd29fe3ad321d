import sys, time
from pathlib import Path
sys.path.insert(0, "/root/repo")
import torch
import flashinfer_amd as fi

def bench_tc_decode(bs, kv, Hq, Hkv, D=128, page=16):
    pages_per = (kv + page - 1) // page
    npages = bs * pages_per
    kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    qo_indptr = torch.arange(0, bs + 1, dtype=torch.int32, device="cuda")
    kv_lens = torch.full((bs,), kv, dtype=torch.int32, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_lens, Hq, Hkv, D, D, page,
           causal=False, q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    fn = lambda: w.run(q, (kc, vc), out=out)
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(20): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 20
    tb = bs * kv * Hkv * D * 2 * 2 / dt / 1e12
    print(f"tc-decode bs={bs} kv={kv} {Hq}q/{Hkv}kv: {dt*1e6:.1f} us  {tb:.2f} TB/s")

bench_tc_decode(16, 1024, 64, 8)
bench_tc_decode(64, 1024, 64, 8)
bench_tc_decode(256, 8192, 64, 8)
bench_tc_decode(256, 32768, 32, 8)
