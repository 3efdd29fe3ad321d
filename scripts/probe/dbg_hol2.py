import math, sys
sys.path.insert(0, "/root/repo")
import torch
import flashinfer_amd as fi

def run_case(qo_lens, kv_lens, tag, force_pf=False):
    torch.manual_seed(0)
    Hq, Hkv, D, page = 64, 8, 128, 16
    pages_per = [(L + page - 1) // page for L in kv_lens]
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)), dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)), dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(sum(qo_lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page, causal=True, q_data_type=torch.bfloat16)
    if force_pf:
        w._group_dec = 0
        # rebuild items as all-prefill
        items = []
        qi = qo_indptr.cpu()
        for r, (ql, kl) in enumerate(zip(qo_lens, kv_lens)):
            for qstart in range(0, max(1, ql * 8), 256):
                for h in range(Hkv):
                    items.append((0, r, qstart, h))
        w._items = torch.tensor(items, dtype=torch.int32, device="cuda").reshape(-1, 4)
        w._n_wgs = min(len(items), 256)
    out, _ = w.run(q, (kc, vc))
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
        print(f"[{tag}] req {b} qo={qo_lens[b]} kv={kv_lens[b]} err={err:.4f}")
        if err > 0.05 and qo_lens[b] == 1:
            o = out[qs].float()[0]; r_ = ref[0][0]
            print("   out[:8]", o[:8].tolist())
            print("   ref[:8]", r_[:8].tolist())
            print("   ratio  ", (o[:8]/r_[:8]).tolist())

run_case([1], [1024], "single-dec")
run_case([1, 1], [1024, 64], "two-dec")
run_case([256, 1], [256, 1024], "pf+dec")
run_case([1], [1024], "single-dec-as-pf", force_pf=True)

def run_variant(tag, n_wgs=None, dec_first=False):
    torch.manual_seed(0)
    Hq, Hkv, D, page = 64, 8, 128, 16
    qo_lens, kv_lens = [256, 1], [256, 1024]
    pages_per = [(L + page - 1) // page for L in kv_lens]
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)), dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)), dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.tensor(kv_lens, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(sum(qo_lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page, causal=True, q_data_type=torch.bfloat16)
    if n_wgs: w._n_wgs = n_wgs
    if dec_first:
        it = w._items.cpu().tolist()
        it.sort(key=lambda r: -r[0])  # kind desc: decode first
        w._items = torch.tensor(it, dtype=torch.int32, device="cuda")
    out, _ = w.run(q, (kc, vc))
    # check only the decode req
    b = 1
    qs = int(qo_indptr[b]); base = int(kv_indptr[b])
    tk, tv = [], []
    for p_ in range(pages_per[b]):
        pg = int(kv_indices[base+p_]); n = min(page, kv_lens[b]-p_*page)
        tk.append(kc[pg,:n]); tv.append(vc[pg,:n])
    kk = torch.cat(tk,0); vv = torch.cat(tv,0)
    g = Hq // Hkv
    logits = torch.einsum("hd,lhd->hl", q[qs].float(), kk.float().repeat_interleave(g,1)) / math.sqrt(D)
    ref = torch.einsum("hl,lhd->hd", torch.softmax(logits,-1), vv.float().repeat_interleave(g,1))
    err = (out[qs].float() - ref).abs().max().item()
    print(f"[{tag}] dec err={err:.4f}")

run_variant("pf+dec n_wgs=1 (serial)", n_wgs=1)
run_variant("pf+dec dec-first", dec_first=True)
run_variant("pf+dec n_wgs=8", n_wgs=8)
