import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from flashinfer_amd._lib import get_ext

torch.manual_seed(0)
dev = "cuda"
Hq, Hkv, D = 32, 8, 128
kv_len = 1
q = torch.randn(Hq, D, dtype=torch.bfloat16, device=dev)
k = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device=dev)
v = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device=dev)

ext = get_ext()
n_items = 1
meta = torch.tensor([0, 0, 0, 1], dtype=torch.int32, device=dev)
work_req, work_chunk, merge_indptr = meta[:1], meta[1:2], meta[2:4]
page_meta = torch.tensor([0, 0, 1, kv_len], dtype=torch.int32, device=dev)
indices, indptr, last = page_meta[:1], page_meta[1:3], page_meta[3:]
tmp_v = torch.full((n_items, Hq, D), 7.0, dtype=torch.float32, device=dev)
tmp_s = torch.full((n_items, Hq), 7.0, dtype=torch.float32, device=dev)
import math
ext.batch_decode_run(q.unsqueeze(0), k.unsqueeze(0), v.unsqueeze(0),
                     indices, indptr, last, 0, work_req, work_chunk, 256,
                     tmp_v, tmp_s, 1/math.sqrt(D), 0.0, -1)
torch.cuda.synchronize()
print("tmp_v[0,0,:8]:", tmp_v[0, 0, :8])
print("tmp_s[0,:8]:", tmp_s[0, :8])
print("expected v[0,0,:8]:", v[0, 0, :8].float())
out = torch.empty(1, Hq, D, dtype=torch.bfloat16, device=dev)
lse = torch.empty(1, Hq, dtype=torch.float32, device=dev)
ext.merge_states(tmp_v, tmp_s, out, lse, merge_indptr, 0, 1)
torch.cuda.synchronize()
print("out[0,0,:8]:", out[0, 0, :8])
print("lse:", lse[0, :8])
# larger case
kv_len = 512
k = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device=dev)
v = torch.randn(kv_len, Hkv, D, dtype=torch.bfloat16, device=dev)
page_meta = torch.tensor([0, 0, 1, kv_len], dtype=torch.int32, device=dev)
indices, indptr, last = page_meta[:1], page_meta[1:3], page_meta[3:]
tmp_v.fill_(7.0); tmp_s.fill_(7.0)
ext.batch_decode_run(q.unsqueeze(0), k.unsqueeze(0), v.unsqueeze(0),
                     indices, indptr, last, 0, work_req, work_chunk, 1024,
                     tmp_v, tmp_s, 1/math.sqrt(D), 0.0, -1)
torch.cuda.synchronize()
print("512 tmp_v[0,0,:4]:", tmp_v[0, 0, :4])
g = Hq // Hkv
kf = k.float().repeat_interleave(g, dim=1)
vf = v.float().repeat_interleave(g, dim=1)
logits = torch.einsum("hd,lhd->hl", q.float(), kf) / math.sqrt(D)
p = torch.softmax(logits, -1)
ref = torch.einsum("hl,lhd->hd", p, vf)
print("512 ref[0,:4]:", ref[0, :4])
