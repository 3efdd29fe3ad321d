import sys
from pathlib import Path
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/scripts")
from perf import bench_decode
bench_decode(bs=256, kv=8192, Hq=64, Hkv=8)
bench_decode(bs=256, kv=8192, Hq=32, Hkv=2)  # GROUP=16

# GDN / SSD chunked prefill throughput
import time
import torch
import flashinfer_amd as fi

def bench_gdn(B=2, L=4096, H=16, D=128):
    cu = torch.arange(0, (B + 1) * L, L, dtype=torch.int32, device="cuda")
    q = torch.randn(B * L, H, D, device="cuda").bfloat16()
    k = torch.nn.functional.normalize(torch.randn(B * L, H, D, device="cuda"), dim=-1).bfloat16()
    v = (torch.randn(B * L, H, D, device="cuda") / 4).bfloat16()
    g = torch.rand(B * L, H, device="cuda") * 0.8 + 0.1
    beta = torch.rand(B * L, H, device="cuda")
    fn = lambda: fi.chunk_gated_delta_rule(q, k, v, g, beta, cu)
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"gdn chunk prefill B={B} L={L} H={H}: {dt*1e3:.2f} ms  {B*L/dt/1e6:.2f} M tok/s")

def bench_ssd(B=4, L=4096, H=8, P=64, N=128, G=8):
    x = (torch.randn(B, L, H, P, device="cuda") / 4).bfloat16()
    dt_ = torch.rand(B, L, H, device="cuda") * 0.5
    A = -torch.rand(H, device="cuda")
    Bm = torch.randn(B, L, G, N, device="cuda").bfloat16() / 4
    Cm = torch.randn(B, L, G, N, device="cuda").bfloat16() / 4
    fn = lambda: fi.mamba_chunk_scan_combined(x, dt_, A, Bm, Cm)
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize()
    d = (time.perf_counter() - t0) / 10
    print(f"ssd scan B={B} L={L} H={H} P={P} N={N}: {d*1e3:.2f} ms  {B*L/d/1e6:.2f} M tok/s")

bench_gdn()
bench_ssd()
