import sys
sys.path.insert(0, "/root/repo")
import torch
import importlib
fmod = importlib.import_module('flashinfer_amd.fused_moe')
from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
from flashinfer_amd.fp8_quantization import per_block_quant_fp8

torch.manual_seed(2)
T, H, inter, E, k = 256, 512, 512, 64, 4
x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 8
w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 8
logits = torch.randn(T, E, device="cuda")
weights, ids = moe_topk_softmax(logits, k)
w13_q, w13_s = per_block_quant_fp8(w13)
w2_q, w2_s = per_block_quant_fp8(w2)
out_flat = fused_moe(x, w13_q, w2_q, weights, ids, w13_scale=w13_s, w2_scale=w2_s)
fm = fmod
# manual z-grid path:
from flashinfer_amd._lib import get_ext
from flashinfer_amd.utils import ceil_div
ext = get_ext()
m_indptr, toc, inv = fm._build_permute(ids, E, align=1)
R = T * k
a_q = torch.empty(R, H, dtype=torch.uint8, device="cuda")
a_s = torch.empty(H // 128, R, dtype=torch.float32, device="cuda")
ext.gather_quant_run(x, toc, a_q, a_s)
h1 = torch.empty(R, 2 * inter, dtype=torch.bfloat16, device="cuda")
ext.gemm_fp8_grouped(a_q, w13_q.view(torch.uint8), h1, m_indptr, None,
                     ceil_div(R, 128) + 1, a_s, w13_s.contiguous(), 1.0, 0)
act_q = torch.empty(R, inter, dtype=torch.uint8, device="cuda")
act_s = torch.empty(inter // 128, R, dtype=torch.float32, device="cuda")
ext.silu_mul_quant_run(h1, act_q, act_s, False)
h2 = torch.empty(R, H, dtype=torch.bfloat16, device="cuda")
ext.gemm_fp8_grouped(act_q, w2_q.view(torch.uint8), h2, m_indptr, None,
                     ceil_div(R, 128) + 1, act_s, w2_s.contiguous(), 1.0, 0)
out_z = torch.empty(T, H, dtype=x.dtype, device="cuda")
ext.moe_finalize(h2, out_z, inv.view(T, k), weights.float().contiguous())
diff = (out_flat.float() - out_z.float()).abs().max().item()
print("flat vs zgrid max diff:", diff)
# torch ref
ref = torch.zeros(T, H, device="cuda")
xf = x.float()
for t in range(16):
    for j in range(k):
        e = int(ids[t, j])
        h1r = xf[t] @ w13[e].float().t()
        act = torch.nn.functional.silu(h1r[:inter]) * h1r[inter:]
        ref[t] += float(weights[t, j]) * (act @ w2[e].float().t())
err_f = (out_flat[:16].float() - ref[:16]).abs().max().item()
err_z = (out_z[:16].float() - ref[:16]).abs().max().item()
print("flat vs ref:", err_f, " zgrid vs ref:", err_z)
print("out scale:", ref[:16].abs().max().item())
