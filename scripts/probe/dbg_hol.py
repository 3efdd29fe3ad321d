import math, sys, pathlib
sys.path.insert(0, "/root/repo")
import torch
import flashinfer_amd as fi

torch.manual_seed(0)
Hq, Hkv, D, page = 64, 8, 128, 16
qo_lens = [512, 1, 1, 300, 1, 128, 1, 1]
kv_lens = [512, 1024, 777, 300, 2048, 128, 64, 1500]
pages_per = [(L + page - 1) // page for L in kv_lens]
qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)), dtype=torch.int32, device="cuda")
kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)), dtype=torch.int32, device="cuda")
npages = int(kv_indptr[-1])
kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
nnz = sum(qo_lens)
q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
w = fi.BatchAttention("NHD")
w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page, causal=True, q_data_type=torch.bfloat16)
print("persistent", w._persistent, "group_dec", w._group_dec, "items", w._items.shape, "n_wgs", w._n_wgs)
out, lse = w.run(q, (kc, vc))
for b in range(len(qo_lens)):
    qs, qe = int(qo_indptr[b]), int(qo_indptr[b+1])
    base = int(kv_indptr[b])
    tk, tv = [], []
    for p_ in range(pages_per[b]):
        pg = int(kv_indices[base+p_]); n = min(page, kv_lens[b]-p_*page)
        tk.append(kc[pg,:n]); tv.append(vc[pg,:n])
    kk = torch.cat(tk,0); vv = torch.cat(tv,0)
    g = Hq // Hkv
    qf = q[qs:qe].float().transpose(0,1)
    kf = kk.float().repeat_interleave(g,dim=1).transpose(0,1)
    vf = vv.float().repeat_interleave(g,dim=1).transpose(0,1)
    logits = qf @ kf.transpose(-1,-2) / math.sqrt(D)
    qpos = torch.arange(qo_lens[b], device="cuda")[:,None]
    kpos = torch.arange(kv_lens[b], device="cuda")[None,:]
    logits = logits.masked_fill((kpos > qpos + (kv_lens[b]-qo_lens[b]))[None], float("-inf"))
    ref = (torch.softmax(logits,-1) @ vf).transpose(0,1)
    err = (out[qs:qe].float() - ref).abs().max().item()
    kind = "dec" if qo_lens[b]==1 else "pf "
    # per-head err for decode reqs
    if err > 0.05 and qo_lens[b]==1:
        perh = (out[qs:qe].float()-ref).abs().amax(dim=(0,2))
        bad = (perh > 0.05).nonzero().flatten().tolist()
        print(f"req {b} [{kind}] kv={kv_lens[b]} maxerr {err:.4f} bad_heads {bad[:16]}")
    else:
        print(f"req {b} [{kind}] kv={kv_lens[b]} maxerr {err:.4f}")
