import sys, importlib
sys.path.insert(0, "/root/repo")
import torch
fm = importlib.import_module('flashinfer_amd.fused_moe')
from flashinfer_amd._lib import get_ext
from flashinfer_amd.utils import ceil_div
from flashinfer_amd.fp8_quantization import per_block_quant_fp8

torch.manual_seed(2)
T, H, inter, E, k = 256, 512, 512, 64, 4
x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 8
logits = torch.randn(T, E, device="cuda")
weights, ids = fm.moe_topk_softmax(logits, k)
w13_q, w13_s = per_block_quant_fp8(w13)
ext = get_ext()

# aligned permute + flat GEMM1
m_indptr, toc, inv = fm._build_permute(ids, E, align=128)
Rp = toc.shape[0]
a_q = torch.empty(Rp, H, dtype=torch.uint8, device="cuda")
a_s = torch.empty(H // 128, Rp, dtype=torch.float32, device="cuda")
ext.gather_quant_run(x, toc, a_q, a_s)
h1_flat = torch.zeros(Rp, 2 * inter, dtype=torch.bfloat16, device="cuda")
ext.gemm_fp8_grouped(a_q, w13_q.view(torch.uint8), h1_flat, m_indptr, None,
                     ceil_div(Rp, 128) + 1, a_s, w13_s.contiguous(), 1.0, Rp // 128)
# same inputs through the Z-GRID kernel (identical math expected)
h1_z = torch.zeros(Rp, 2 * inter, dtype=torch.bfloat16, device="cuda")
ext.gemm_fp8_grouped(a_q, w13_q.view(torch.uint8), h1_z, m_indptr, None,
                     ceil_div(Rp, 128) + 1, a_s, w13_s.contiguous(), 1.0, 0)
mi = m_indptr.cpu().tolist()
cnt = torch.diff(torch.tensor(mi)).tolist()
bad = []
for e in range(E):
    s0, s1 = mi[e], mi[e + 1]
    if s0 == s1: continue
    d = (h1_flat[s0:s1].float() - h1_z[s0:s1].float()).abs().max().item()
    if d > 0.05: bad.append((e, s0, s1, round(d, 3)))
print("m_indptr head:", mi[:10], "... total", mi[-1], "Rp", Rp)
print("bad segs:", bad[:10], " nbad:", len(bad))

# determinism + pattern analysis
h1_flat2 = torch.zeros_like(h1_flat)
ext.gemm_fp8_grouped(a_q, w13_q.view(torch.uint8), h1_flat2, m_indptr, None,
                     ceil_div(Rp, 128) + 1, a_s, w13_s.contiguous(), 1.0, Rp // 128)
print("flat deterministic:", torch.equal(h1_flat, h1_flat2))
e, s0, s1 = bad[0][0], bad[0][1], bad[0][2]
d = (h1_flat[s0:s1].float() - h1_z[s0:s1].float()).abs()
rowmax = d.amax(1)
colmax = d.amax(0)
print("seg", e, "bad rows:", (rowmax > 0.05).sum().item(), "/128",
      "first bad rows:", (rowmax > 0.05).nonzero().flatten()[:8].tolist())
print("bad cols:", (colmax > 0.05).sum().item(), "/", d.shape[1],
      "first bad cols:", (colmax > 0.05).nonzero().flatten()[:8].tolist())
