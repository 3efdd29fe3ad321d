"""Round-2 PMC capture target: flagship kernels in tight loops."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent.parent))
import torch
import flashinfer_amd as fi

which = sys.argv[1]
torch.manual_seed(0)
if which == "prefill":
    bs, s, Hq, Hkv, D, page = 16, 1024, 32, 8, 128, 16
    pp = s // page
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    idx = torch.randperm(bs * pp, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = torch.randn(bs * pp, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(bs * pp, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs * s, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, idx, lpl, Hq, Hkv, D, page, causal=True)
    out = torch.empty_like(q)
    for _ in range(50):
        w.run(q, (kc, vc), out=out)
elif which == "decode_mfma":
    bs, kv, Hq, Hkv, D, page = 16, 1024, 64, 8, 128, 16
    pp = kv // page
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    idx = torch.randperm(bs * pp, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), page, dtype=torch.int32, device="cuda")
    kc = torch.randn(bs * pp, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(bs * pp, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(kv_indptr, idx, lpl, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    for _ in range(200):
        w.run(q, (kc, vc), out=out)
torch.cuda.synchronize()
print("done", which)
