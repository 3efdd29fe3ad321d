import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
import flashinfer_amd as fi

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

# MoE shapes: Mixtral 8x7B w13: E=8, N=2*14336, K=4096, T=4096 top2 -> 8192 rows
E, N, K, rows = 8, 28672, 4096, 8192
per = rows // E
a = torch.randn(rows, K, dtype=torch.bfloat16, device="cuda")
b = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") / 16
m_indptr = torch.arange(0, rows + 1, per, dtype=torch.int32, device="cuda")
t_us = bench(lambda: fi.grouped_mm_bf16(a, b, m_indptr))
fl = 2 * rows * N * K
print(f"in-house grouped: {fl/t_us/1e12:8.1f} TF ({t_us*1e6:.0f}us)")
try:
    offs = m_indptr[1:].to(torch.int32)
    t_tg = bench(lambda: torch._grouped_mm(a, b.transpose(1, 2), offs=offs))
    print(f"torch._grouped_mm: {fl/t_tg/1e12:8.1f} TF ({t_tg*1e6:.0f}us)")
    r1 = fi.grouped_mm_bf16(a, b, m_indptr)
    r2 = torch._grouped_mm(a, b.transpose(1, 2), offs=offs)
    print("match:", torch.allclose(r1.float(), r2.float(), atol=2e-1, rtol=2e-2))
except Exception as e:
    print("torch._grouped_mm failed:", type(e).__name__, str(e)[:200])

# empty-group behavior
m2 = torch.tensor([0, 0, 4096, 4096, 8192, 8192, 8192, 8192, 8192],
                  dtype=torch.int32, device="cuda")
try:
    r = torch._grouped_mm(a, b.transpose(1, 2), offs=m2[1:])
    ref = fi.grouped_mm_bf16(a, b, m2)
    print("empty-group match:", torch.allclose(r.float(), ref.float(),
                                               atol=2e-1, rtol=2e-2))
except Exception as e:
    print("empty-group failed:", type(e).__name__, str(e)[:150])
