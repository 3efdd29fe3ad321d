"""Benchmark framework (reference parity: benchmarks/flashinfer_benchmark.py
+ routines/ — per-routine perf measurement with --refcheck output
verification and testlist batch runs).

Usage:
  python benchmarks/flashinfer_benchmark.py --routine batch_prefill \
      --batch 16 --s 1024 --refcheck
  python benchmarks/flashinfer_benchmark.py --testlist benchmarks/sample_testlist.txt
Outputs one CSV row per run:
  routine,config,median_us,metric,value,refcheck
"""
from __future__ import annotations

import argparse
import math
import shlex
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def _median_time(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    times = []
    for _ in range(iters):
        e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
        e0.record()
        fn()
        e1.record()
        torch.cuda.synchronize()
        times.append(e0.elapsed_time(e1) * 1e3)  # us
    times.sort()
    return times[len(times) // 2]


def routine_batch_prefill(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    bs, s, Hq, Hkv, D, page = a.batch, a.s, a.num_qo_heads, a.num_kv_heads, a.head_dim, a.page_size
    pp = (s + page - 1) // page
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    npages = bs * pp
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), (s - 1) % page + 1, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs * s, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, last, Hq, Hkv, D, page, causal=True)
    out = torch.empty_like(q)
    t = _median_time(lambda: w.run(q, (kc, vc), out=out))
    fl = bs * Hq * D * 4 * (s * (s + 1) / 2)
    ok = ""
    if a.refcheck:
        o = w.run(q, (kc, vc))
        ref = _sdpa_paged_ref(q[: s], kc, vc, kv_indices[:pp], s, page, causal=True)
        ok = "PASS" if torch.allclose(o[:s].float(), ref, atol=3e-2, rtol=3e-2) else "FAIL"
    return t, "TFLOPS", fl / (t * 1e-6) / 1e12, ok


def _sdpa_paged_ref(q, kc, vc, page_ids, s, page, causal):
    rows_k = torch.cat([kc[int(p)] for p in page_ids])[:s]
    rows_v = torch.cat([vc[int(p)] for p in page_ids])[:s]
    Hq, D = q.shape[1], q.shape[2]
    g = Hq // rows_k.shape[1]
    kf = rows_k.float().repeat_interleave(g, 1)
    vf = rows_v.float().repeat_interleave(g, 1)
    logits = torch.einsum("mhd,lhd->hml", q.float(), kf) / math.sqrt(D)
    if causal:
        qp = torch.arange(s, device="cuda")[:, None]
        kp = torch.arange(s, device="cuda")[None, :]
        logits = logits.masked_fill((kp > qp)[None], float("-inf"))
    return torch.einsum("hml,lhd->mhd", torch.softmax(logits, -1), vf)


def routine_batch_decode(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    bs, kv, Hq, Hkv, D, page = a.batch, a.s_kv, a.num_qo_heads, a.num_kv_heads, a.head_dim, a.page_size
    pp = (kv + page - 1) // page
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    npages = bs * pp
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32, device="cuda")
    kc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(kv_indptr, kv_indices, last, Hq, Hkv, D, page,
           q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    t = _median_time(lambda: w.run(q, (kc, vc), out=out))
    tb = bs * kv * Hkv * D * 4 / (t * 1e-6) / 1e12
    ok = ""
    if a.refcheck:
        o = w.run(q, (kc, vc))
        from tests.test_decode import sdpa_ref

        rows_k = torch.cat([kc[int(p)] for p in kv_indices[:pp]])[:kv]
        rows_v = torch.cat([vc[int(p)] for p in kv_indices[:pp]])[:kv]
        ref = sdpa_ref(q[0], rows_k, rows_v)
        ok = "PASS" if torch.allclose(o[0].float(), ref, atol=3e-2, rtol=3e-2) else "FAIL"
    return t, "TB/s", tb, ok


def routine_gemm_bf16(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    M = N = K = a.n
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    wt = torch.randn(N, K, dtype=torch.bfloat16, device="cuda").t()  # [K, N] col-major
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    t = _median_time(lambda: fi.mm_bf16(x, wt, out=out), iters=10)
    ok = ""
    if a.refcheck:
        o = fi.mm_bf16(x, wt)
        ref = x.float() @ wt.float()
        ok = "PASS" if torch.allclose(o.float(), ref, atol=2.0, rtol=3e-2) else "FAIL"
    return t, "TFLOPS", 2 * M * N * K / (t * 1e-6) / 1e12, ok


def routine_mla_decode(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    bs, kv, H, page = a.batch, a.s_kv, 128, 32
    pp = (kv + page - 1) // page
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    npages = bs * pp
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.full((bs,), kv, dtype=torch.int32, device="cuda")
    qo_indptr = torch.arange(0, bs + 1, dtype=torch.int32, device="cuda")
    ckv = torch.randn(npages, page, 512, dtype=torch.bfloat16, device="cuda")
    kpe = torch.randn(npages, page, 64, dtype=torch.bfloat16, device="cuda")
    qn = torch.randn(bs, H, 512, dtype=torch.bfloat16, device="cuda")
    qp = torch.randn(bs, H, 64, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(2048 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, H, 512, 64, page, True,
           1 / math.sqrt(576), torch.bfloat16)
    out = torch.empty(bs, H, 512, dtype=torch.bfloat16, device="cuda")
    t = _median_time(lambda: w.run(qn, qp, ckv, kpe, out=out))
    fl = 2 * bs * H * kv * (576 + 512)
    return t, "TFLOPS", fl / (t * 1e-6) / 1e12, ""


def routine_batch_prefill_ragged(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    bs, s, Hq, Hkv = a.batch, a.s, a.num_qo_heads, a.num_kv_heads
    Dqk, Dvo = a.head_dim, a.head_dim_vo or a.head_dim
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    q = torch.randn(bs * s, Hq, Dqk, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(bs * s, Hkv, Dqk, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(bs * s, Hkv, Dvo, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, qo_indptr.clone(), Hq, Hkv, Dqk, head_dim_vo=Dvo,
           causal=True, q_data_type=torch.bfloat16)
    out = torch.empty(bs * s, Hq, Dvo, dtype=torch.bfloat16, device="cuda")
    t = _median_time(lambda: w.run(q, k, v, out=out))
    fl = bs * Hq * (Dqk + Dvo) * (s * (s + 1) / 2) * 2
    ok = ""
    if a.refcheck:
        from tests.test_prefill import ref_attn_vo

        o = w.run(q, k, v)
        ref = ref_attn_vo(q[:s], k[:s], v[:s], causal=True)
        ok = "PASS" if torch.allclose(o[:s].float(), ref, atol=3e-2, rtol=3e-2) else "FAIL"
    return t, "TFLOPS", fl / (t * 1e-6) / 1e12, ok


def routine_gemm_fp8_groupwise(a):
    from flashinfer_amd.fp8_quantization import (
        gemm_fp8_nt_groupwise, per_block_quant_fp8, per_token_group_quant_fp8)

    torch.manual_seed(0)
    M = N = K = a.n
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    wt = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    a_q, a_s = per_token_group_quant_fp8(x, transpose_scale=True)
    b_q, b_s = per_block_quant_fp8(wt)
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    t = _median_time(lambda: gemm_fp8_nt_groupwise(a_q, b_q, a_s, b_s, out=out),
                     iters=10)
    ok = ""
    if a.refcheck:
        o = gemm_fp8_nt_groupwise(a_q, b_q, a_s, b_s)
        ref = x.float() @ wt.float().t()
        rel = (o.float() - ref).abs().mean() / ref.abs().mean()
        ok = "PASS" if float(rel) < 0.05 else "FAIL"
    return t, "TFLOPS", 2 * M * N * K / (t * 1e-6) / 1e12, ok


def routine_fused_moe(a):
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import (per_block_quant_fp8,
                                                 per_block_quant_mxfp8)

    torch.manual_seed(0)
    T, H, inter, E, k = a.batch, a.n, a.inter, a.experts, a.top_k
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 16
    weights, ids = moe_topk_softmax(torch.randn(T, E, device="cuda"), k)
    kw = {}
    if a.dtype == "fp8":
        w13, w13_s = per_block_quant_fp8(w13)
        w2, w2_s = per_block_quant_fp8(w2)
        kw = dict(w13_scale=w13_s, w2_scale=w2_s)
    elif a.dtype == "mxfp8":
        w13, w13_s = per_block_quant_mxfp8(w13)
        w2, w2_s = per_block_quant_mxfp8(w2)
        kw = dict(w13_scale=w13_s, w2_scale=w2_s)
    t = _median_time(lambda: fused_moe(x, w13, w2, weights, ids, **kw), iters=10)
    fl = T * k * 3 * H * inter * 2
    return t, "TFLOPS", fl / (t * 1e-6) / 1e12, ""


def routine_rmsnorm(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(a.batch, a.n, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(a.n, dtype=torch.bfloat16, device="cuda")
    t = _median_time(lambda: fi.rmsnorm(x, w))
    gb = 2 * x.numel() * 2 / (t * 1e-6) / 1e9
    ok = ""
    if a.refcheck:
        o = fi.rmsnorm(x, w)
        xf = x.float()
        ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
        ok = "PASS" if torch.allclose(o.float(), ref, atol=2e-2, rtol=2e-2) else "FAIL"
    return t, "GB/s", gb, ok


def routine_rope(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    nnz, Hq, Hkv, D = a.batch * a.s, a.num_qo_heads, a.num_kv_heads, a.head_dim
    q = torch.randn(nnz, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    indptr = torch.arange(0, (a.batch + 1) * a.s, a.s, dtype=torch.int32, device="cuda")
    offsets = torch.zeros(a.batch, dtype=torch.int32, device="cuda")
    t = _median_time(lambda: fi.apply_rope(q, k, indptr, offsets))
    gb = 2 * (q.numel() + k.numel()) * 2 / (t * 1e-6) / 1e9
    return t, "GB/s", gb, ""


def routine_sampling_topk(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    probs = torch.softmax(torch.randn(a.batch, a.n, device="cuda"), -1)
    t = _median_time(lambda: fi.top_k_sampling_from_probs(probs, 50))
    return t, "Mrows/s", a.batch / (t * 1e-6) / 1e6, ""


def routine_moe_routing(a):
    from flashinfer_amd.fused_moe import dsv3_routing

    torch.manual_seed(0)
    logits = torch.randn(a.batch, a.experts, device="cuda")
    bias = torch.randn(a.experts, device="cuda") * 0.1
    t = _median_time(lambda: dsv3_routing(logits, a.top_k, 8, 4, 2.5, bias))
    return t, "Mtok/s", a.batch / (t * 1e-6) / 1e6, ""


def routine_append_kv(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    bs, s, Hkv, D, page = a.batch, a.s, a.num_kv_heads, a.head_dim, a.page_size
    nnz = bs * s
    pp = (s + page - 1) // page
    npages = bs * pp
    kc = torch.zeros(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(nnz, Hkv, D, dtype=torch.bfloat16, device="cuda")
    append_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    kv_indptr = torch.arange(0, (bs + 1) * pp, pp, dtype=torch.int32, device="cuda")
    kv_indices = torch.arange(npages, dtype=torch.int32, device="cuda")
    lpl = torch.full((bs,), (s - 1) % page + 1, dtype=torch.int32, device="cuda")
    seq_lens = fi.get_seq_lens(kv_indptr, lpl, page)
    bi, pos = fi.get_batch_indices_positions(append_indptr, seq_lens, nnz)
    t = _median_time(lambda: fi.append_paged_kv_cache(
        k, v, bi, pos, (kc, vc), kv_indices, kv_indptr, lpl))
    gb = 2 * (k.numel() + v.numel()) * 2 / (t * 1e-6) / 1e9
    return t, "GB/s", gb, ""


def routine_topk(a):
    import flashinfer_amd as fi

    torch.manual_seed(0)
    x = torch.randn(a.batch, a.n, device="cuda")
    t = _median_time(lambda: fi.top_k(x, 256))
    ok = ""
    if a.refcheck:
        vals, _ = fi.top_k(x, 256)
        ref = torch.topk(x, 256, dim=-1).values
        ok = ("PASS" if torch.allclose(vals.sort(-1, descending=True).values,
                                       ref, atol=1e-5) else "FAIL")
    return t, "Mrows/s", a.batch / (t * 1e-6) / 1e6, ok



ROUTINES = {
    "batch_prefill": routine_batch_prefill,
    "batch_prefill_ragged": routine_batch_prefill_ragged,
    "batch_decode": routine_batch_decode,
    "gemm_bf16": routine_gemm_bf16,
    "gemm_fp8_groupwise": routine_gemm_fp8_groupwise,
    "fused_moe": routine_fused_moe,
    "mla_decode": routine_mla_decode,
    "rmsnorm": routine_rmsnorm,
    "rope": routine_rope,
    "sampling_topk": routine_sampling_topk,
    "moe_routing": routine_moe_routing,
    "append_kv": routine_append_kv,
    "topk": routine_topk,
}


def run_one(argv):
    ap = argparse.ArgumentParser()
    ap.add_argument("--routine", required=True, choices=sorted(ROUTINES))
    ap.add_argument("--batch", type=int, default=16)
    ap.add_argument("--s", type=int, default=1024)
    ap.add_argument("--s_kv", type=int, default=1024)
    ap.add_argument("--num_qo_heads", type=int, default=32)
    ap.add_argument("--num_kv_heads", type=int, default=8)
    ap.add_argument("--head_dim", type=int, default=128)
    ap.add_argument("--page_size", type=int, default=16)
    ap.add_argument("--n", type=int, default=4096)
    ap.add_argument("--head_dim_vo", type=int, default=0)
    ap.add_argument("--inter", type=int, default=14336)
    ap.add_argument("--experts", type=int, default=8)
    ap.add_argument("--top_k", type=int, default=2)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp8", "mxfp8"])
    ap.add_argument("--refcheck", action="store_true")
    a = ap.parse_args(argv)
    t, metric, value, ok = ROUTINES[a.routine](a)
    cfg = " ".join(argv)
    print(f"{a.routine},{cfg!r},{t:.1f},{metric},{value:.1f},{ok}")


def main():
    if "--testlist" in sys.argv:
        path = sys.argv[sys.argv.index("--testlist") + 1]
        print("routine,config,median_us,metric,value,refcheck")
        for line in Path(path).read_text().splitlines():
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            run_one(shlex.split(line))
    else:
        run_one(sys.argv[1:])


if __name__ == "__main__":
    main()
