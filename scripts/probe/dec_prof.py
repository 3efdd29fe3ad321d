"""Focused profiling target: one decode config in a tight loop (for rocprofv3)."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent.parent))
import torch
import flashinfer_amd as fi

mode = sys.argv[1] if len(sys.argv) > 1 else "mfma"
bs = int(sys.argv[2]) if len(sys.argv) > 2 else 16
kv = int(sys.argv[3]) if len(sys.argv) > 3 else 1024
Hq, Hkv, D, page = 64, 8, 128, 16
torch.manual_seed(0)
pages_per = (kv + page - 1) // page
kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
npages = bs * pages_per
kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
last_page = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32, device="cuda")
kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
kwargs = {}
if mode == "tc":
    kwargs = dict(use_tensor_cores=True)
w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", **kwargs)
w.plan(kv_indptr, kv_indices, last_page, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
out = torch.empty_like(q)
for _ in range(200):
    w.run(q, (kc, vc), out=out)
torch.cuda.synchronize()
print("done", mode, "fused_mfma" if getattr(w, "_fused_mfma", False) else ("tc" if w._tc else "other"))
