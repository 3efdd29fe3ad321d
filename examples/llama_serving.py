"""End-to-end serving example: a Llama-3-8B-shaped transformer layer stack
running entirely on flashinfer_amd kernels — the integration pattern a
serving engine (vLLM/SGLang-style) would use.

Per step:
  prefill:  rmsnorm -> qkv GEMM -> rope -> append to paged KV ->
            BatchPrefillWithPagedKVCacheWrapper -> o GEMM -> residual ->
            rmsnorm -> MLP (GEMM + silu_and_mul + GEMM) -> residual
  decode:   same with BatchDecodeWithPagedKVCacheWrapper
  sampling: top-k/top-p rejection sampling over the LM head logits

Run on an MI355X:  python examples/llama_serving.py
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import math

import torch

import flashinfer_amd as fi


class LlamaLayer:
    def __init__(self, hidden=4096, inter=14336, Hq=32, Hkv=8, D=128,
                 dtype=torch.bfloat16, dev="cuda"):
        def w(n, k):
            return (torch.randn(n, k, dtype=dtype, device=dev) / math.sqrt(k)).t()

        self.hidden, self.Hq, self.Hkv, self.D = hidden, Hq, Hkv, D
        self.wq = w(Hq * D, hidden)      # [hidden, Hq*D] column-major
        self.wk = w(Hkv * D, hidden)
        self.wv = w(Hkv * D, hidden)
        self.wo = w(hidden, Hq * D)
        self.w13 = w(2 * inter, hidden)
        self.w2 = w(hidden, inter)
        self.norm1 = torch.randn(hidden, dtype=dtype, device=dev)
        self.norm2 = torch.randn(hidden, dtype=dtype, device=dev)

    def forward(self, x, attn_run, pos_ids, append_fn):
        h = fi.rmsnorm(x, self.norm1)
        q = fi.mm_bf16(h, self.wq).view(-1, self.Hq, self.D)
        k = fi.mm_bf16(h, self.wk).view(-1, self.Hkv, self.D)
        v = fi.mm_bf16(h, self.wv).view(-1, self.Hkv, self.D)
        q, k = fi.apply_rope_pos_ids(q, k, pos_ids)
        append_fn(k, v)
        attn = attn_run(q).reshape(-1, self.Hq * self.D)
        x = x + fi.mm_bf16(attn, self.wo)
        h = fi.rmsnorm(x, self.norm2)
        h = fi.silu_and_mul(fi.mm_bf16(h, self.w13))
        return x + fi.mm_bf16(h, self.w2)


def main(batch=4, prompt_len=128, gen_tokens=8, layers=2, vocab=32000):
    torch.manual_seed(0)
    dev = "cuda"
    hidden, Hq, Hkv, D, page = 4096, 32, 8, 128, 16
    model = [LlamaLayer(hidden=hidden, Hq=Hq, Hkv=Hkv, D=D, dev=dev)
             for _ in range(layers)]
    embed = torch.randn(vocab, hidden, dtype=torch.bfloat16, device=dev) / 64
    lm_head = (torch.randn(vocab, hidden, dtype=torch.bfloat16, device=dev)
               / math.sqrt(hidden)).t()

    # paged KV cache per layer
    max_len = prompt_len + gen_tokens
    pages_per = (max_len + page - 1) // page
    npages = batch * pages_per
    caches = [
        (torch.zeros(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev),
         torch.zeros(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev))
        for _ in range(layers)
    ]
    kv_indptr = torch.arange(0, (batch + 1) * pages_per, pages_per,
                             dtype=torch.int32, device=dev)
    kv_indices = torch.arange(npages, dtype=torch.int32, device=dev)

    ws = torch.empty(256 << 20, dtype=torch.uint8, device=dev)
    prefill = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    ws2 = torch.empty(256 << 20, dtype=torch.uint8, device=dev)
    decode = fi.BatchDecodeWithPagedKVCacheWrapper(ws2, "NHD")
    gen = torch.Generator(device=dev).manual_seed(7)

    tokens = torch.randint(0, vocab, (batch, prompt_len), device=dev)
    seq_lens = torch.full((batch,), prompt_len, dtype=torch.int32, device=dev)

    def lens_state(extra=0):
        lens = seq_lens + extra
        last = ((lens - 1) % page + 1).to(torch.int32)
        return lens, last

    # ---- prefill ----
    lens, last = lens_state()
    qo_indptr = torch.arange(0, (batch + 1) * prompt_len, prompt_len,
                             dtype=torch.int32, device=dev)
    prefill.plan(qo_indptr, kv_indptr, kv_indices, last, Hq, Hkv, D, page,
                 causal=True)
    bi, pos = fi.get_batch_indices_positions(qo_indptr, lens, batch * prompt_len)
    x = embed[tokens.reshape(-1)]
    for li, layer in enumerate(model):
        def append(k, v, li=li):
            fi.append_paged_kv_cache(k, v, bi, pos, caches[li], kv_indices,
                                     kv_indptr, last, "NHD")
        x = layer.forward(
            x, lambda q: prefill.run(q, caches[li]), pos, append)
    logits = fi.mm_bf16(x[prompt_len - 1::prompt_len], lm_head).float()
    next_tok = fi.top_k_top_p_sampling_from_probs(
        fi.softmax(logits, temperature=0.8), 40, 0.95, generator=gen)
    out_tokens = [next_tok]

    # ---- decode loop ----
    for step in range(gen_tokens - 1):
        lens, last = lens_state(step + 1)
        decode.plan(kv_indptr, kv_indices, last, Hq, Hkv, D, page,
                    q_data_type=torch.bfloat16)
        dec_indptr = torch.arange(0, batch + 1, dtype=torch.int32, device=dev)
        bi_d, pos_d = fi.get_batch_indices_positions(dec_indptr, lens, batch)
        x = embed[next_tok.long()]
        pos_ids = (lens - 1).to(torch.int32)
        for li, layer in enumerate(model):
            def append(k, v, li=li):
                fi.append_paged_kv_cache(k, v, bi_d, pos_d, caches[li],
                                         kv_indices, kv_indptr, last, "NHD")
            x = layer.forward(
                x, lambda q: decode.run(q, caches[li]), pos_ids, append)
        logits = fi.mm_bf16(x, lm_head).float()
        next_tok = fi.top_k_top_p_sampling_from_probs(
            fi.softmax(logits, temperature=0.8), 40, 0.95, generator=gen)
        out_tokens.append(next_tok)

    result = torch.stack(out_tokens, dim=1)
    assert result.shape == (batch, gen_tokens)
    assert (result >= 0).all() and (result < vocab).all()
    print(f"generated {gen_tokens} tokens for {batch} requests: OK")
    print(result.cpu().tolist())
    return result


if __name__ == "__main__":
    main()
