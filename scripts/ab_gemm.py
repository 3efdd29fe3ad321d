import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
import flashinfer_amd as fi

def bench(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for (M, N, K) in [(8192,8192,8192),(4096,4096,4096),(1024,14336,4096),
                  (256,4096,4096),(256,14336,4096),(64,4096,4096),(8,4096,4096)]:
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, K, dtype=torch.bfloat16, device="cuda").t()
    t_lib = bench(lambda: torch.mm(a, b))
    t_us = bench(lambda: fi.mm_bf16(a, b, backend="mfma"))
    fl = 2 * M * N * K
    print(f"M{M} N{N} K{K}: hipblaslt {fl/t_lib/1e12:8.1f} TF ({t_lib*1e6:8.1f}us)"
          f"  mfma {fl/t_us/1e12:8.1f} TF ({t_us*1e6:8.1f}us)")

# fp8 per-tensor: our kernel vs hipBLASLt (torch._scaled_mm)
print("\nfp8 e4m3 per-tensor:")
for (M, N, K) in [(8192,8192,8192),(4096,4096,4096),(1024,14336,4096),(256,4096,4096)]:
    a = (torch.randn(M, K, device="cuda") / 8).to(torch.float8_e4m3fn)
    b = (torch.randn(N, K, device="cuda") / 8).to(torch.float8_e4m3fn).t()
    sa = torch.tensor(1.0, device="cuda"); sb = torch.tensor(1.0, device="cuda")
    try:
        t_lib = bench(lambda: torch._scaled_mm(a, b, scale_a=sa, scale_b=sb,
                                               out_dtype=torch.bfloat16))
    except Exception as e:
        print("  _scaled_mm failed:", e); break
    t_us = bench(lambda: fi.mm_fp8(a, b, sa, sb))
    fl = 2 * M * N * K
    print(f"M{M} N{N} K{K}: hipblaslt {fl/t_lib/1e12:8.1f} TF  ours {fl/t_us/1e12:8.1f} TF")
