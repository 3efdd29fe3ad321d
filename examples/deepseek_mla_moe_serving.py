"""DeepSeek-V3-style serving slice on flashinfer_amd: MLA paged decode
(compressed KV 512 + rope 64, matrix-absorbed) + no-aux-loss routed fp8 MoE,
assembled from the same kernels a DSv3 engine would use.

Per decode step and layer:
  rmsnorm -> q/kv down-projections -> concat_mla_k-style cache append ->
  BatchMLAPagedAttentionWrapper.run (fp8 ckv/kpe cache, dequant-in-LDS) ->
  o projection -> rmsnorm -> dsv3_routing (grouped sigmoid top-k) ->
  fused_moe fp8 (1x128 activation x 128x128 weight groupwise scales)

Run on an MI355X:  python examples/deepseek_mla_moe_serving.py
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import math

import torch

import flashinfer_amd as fi


def main(batch=8, ctx_len=512, steps=4, layers=2):
    torch.manual_seed(0)
    dev = "cuda"
    H = 16            # attention heads (scaled-down DSv3)
    D_CKV, D_KPE = 512, 64
    hidden = 1024
    E, top_k, n_group, topk_group = 16, 4, 4, 2
    inter = 512
    page = 16

    # paged fp8 MLA cache
    max_len = ctx_len + steps
    pages_per = (max_len + page - 1) // page
    npages = batch * pages_per
    ckv_cache = (torch.randn(npages, page, D_CKV, device=dev) / 8).to(
        torch.float8_e4m3fn)
    kpe_cache = (torch.randn(npages, page, D_KPE, device=dev) / 8).to(
        torch.float8_e4m3fn)
    kv_indptr = torch.arange(0, (batch + 1) * pages_per, pages_per,
                             dtype=torch.int32, device=dev)
    kv_indices = torch.arange(npages, dtype=torch.int32, device=dev)

    def w(n, k):
        return (torch.randn(n, k, dtype=torch.bfloat16, device=dev)
                / math.sqrt(k)).t()

    class Layer:
        def __init__(self):
            self.norm1 = torch.randn(hidden, dtype=torch.bfloat16, device=dev)
            self.norm2 = torch.randn(hidden, dtype=torch.bfloat16, device=dev)
            self.wq_nope = w(H * D_CKV, hidden)
            self.wq_pe = w(H * D_KPE, hidden)
            self.wkv = w(D_CKV + D_KPE, hidden)
            self.wo = w(hidden, H * D_CKV)
            self.router = torch.randn(E, hidden, dtype=torch.bfloat16,
                                      device=dev) / math.sqrt(hidden)
            w13, s13 = fi.per_block_quant_fp8(
                torch.randn(E, 2 * inter, hidden, device=dev) / math.sqrt(hidden))
            w2, s2 = fi.per_block_quant_fp8(
                torch.randn(E, hidden, inter, device=dev) / math.sqrt(inter))
            self.w13, self.s13, self.w2, self.s2 = w13, s13, w2, s2

    model = [Layer() for _ in range(layers)]
    ws = torch.empty(256 << 20, dtype=torch.uint8, device=dev)
    mla = fi.BatchMLAPagedAttentionWrapper(ws)
    x = torch.randn(batch, hidden, dtype=torch.bfloat16, device=dev)
    lens = torch.full((batch,), ctx_len, dtype=torch.int32, device=dev)
    qo_indptr = torch.arange(0, batch + 1, dtype=torch.int32, device=dev)
    bi = torch.arange(batch, dtype=torch.int32, device=dev)

    for step in range(steps):
        lens += 1
        last = ((lens - 1) % page + 1).int()
        mla.plan(qo_indptr, kv_indptr, kv_indices, lens, H, D_CKV, D_KPE,
                 page, causal=False, sm_scale=(D_CKV + D_KPE) ** -0.5,
                 q_data_type=torch.bfloat16)
        pos = (lens - 1).int()
        for layer in model:
            h = fi.rmsnorm(x, layer.norm1)
            q_nope = fi.mm_bf16(h, layer.wq_nope).view(batch, H, D_CKV)
            q_pe = fi.mm_bf16(h, layer.wq_pe).view(batch, H, D_KPE)
            kv = fi.mm_bf16(h, layer.wkv)
            fi.append_paged_mla_kv_cache(
                kv[:, :D_CKV], kv[:, D_CKV:], bi, pos, ckv_cache, kpe_cache,
                kv_indices, kv_indptr, last)
            attn = mla.run(q_nope, q_pe, ckv_cache, kpe_cache,
                           ckv_scale=1.0, kpe_scale=1.0)
            x = x + fi.mm_bf16(attn.reshape(batch, H * D_CKV), layer.wo)
            h = fi.rmsnorm(x, layer.norm2)
            logits = fi.mm_bf16(h, layer.router.t())
            weights, ids = fi.dsv3_routing(logits.float(), top_k, n_group,
                                           topk_group, 2.5)
            moe_out = fi.fused_moe(h, layer.w13, layer.w2, weights, ids,
                                   w13_scale=layer.s13, w2_scale=layer.s2)
            x = x + moe_out.to(x.dtype)
        assert x.isfinite().all(), f"non-finite at step {step}"
    torch.cuda.synchronize()
    print(f"DSv3 slice: {steps} decode steps x {layers} layers "
          f"(MLA fp8 cache + routed fp8 MoE): OK")


if __name__ == "__main__":
    main()
