"""Quick perf sweep: prefill TFLOPS, decode tok/s+TB/s, GEMM TFLOPS."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import math
import torch
import flashinfer_amd as fi


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1e-3  # seconds


def bench_prefill(bs=16, s=1024, Hq=32, Hkv=8, D=128, page=16, causal=True):
    torch.manual_seed(0)
    qo_lens = [s] * bs
    kv_lens = [s] * bs
    qo_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(qo_lens), 0)), dtype=torch.int32, device="cuda")
    pages_per = [(L + page - 1) // page for L in kv_lens]
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)), dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last_page = torch.tensor([(L - 1) % page + 1 for L in kv_lens], dtype=torch.int32, device="cuda")
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(sum(qo_lens), Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, last_page, Hq, Hkv, D, page, causal=causal)
    out = torch.empty_like(q)
    t = timeit(lambda: w.run(q, (k_cache, v_cache), out=out))
    # flops: per request: Hq * qo * kv * D * 2 (QK) * 2 (PV) [causal: ~half]
    fl = 0
    for qo, kv in zip(qo_lens, kv_lens):
        full = 2 * 2 * Hq * D * qo * kv
        fl += full / 2 + 2 * 2 * Hq * D * qo / 2 if causal else full
    print(f"prefill bs={bs} s={s} causal={causal}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TFLOPS")


def bench_decode(bs=256, kv=32768, Hq=32, Hkv=8, D=128, page=16):
    torch.manual_seed(0)
    kv_lens = [kv] * bs
    pages_per = [(L + page - 1) // page for L in kv_lens]
    kv_indptr = torch.tensor([0] + list(torch.cumsum(torch.tensor(pages_per), 0)), dtype=torch.int32, device="cuda")
    npages = int(kv_indptr[-1])
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last_page = torch.tensor([(L - 1) % page + 1 for L in kv_lens], dtype=torch.int32, device="cuda")
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(512 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(kv_indptr, kv_indices, last_page, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    t = timeit(lambda: w.run(q, (k_cache, v_cache), out=out))
    bytes_kv = bs * kv * Hkv * D * 2 * 2
    print(f"decode bs={bs} kv={kv}: {t*1e6:.1f} us  {bs/t:.0f} tok/s  {bytes_kv/t/1e12:.2f} TB/s")


def bench_gemm(N=4096):
    torch.manual_seed(0)
    a = torch.randn(N, N, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, N, dtype=torch.bfloat16, device="cuda").t().contiguous().t()
    c = torch.empty(N, N, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: fi.mm_bf16(a, b, out=c), iters=10)
    print(f"gemm {N}^3: {t*1e3:.2f} ms  {2*N**3/t/1e12:.0f} TFLOPS")
    # torch (hipBLASLt) comparison
    t2 = timeit(lambda: torch.matmul(a, b, out=c), iters=10)
    print(f"torch {N}^3: {t2*1e3:.2f} ms  {2*N**3/t2/1e12:.0f} TFLOPS")


def bench_mla(bs=16, kv=1024, H=128, page=32):
    import math
    torch.manual_seed(0)
    pages_per = (kv + page - 1) // page
    kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
    npages = bs * pages_per
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.full((bs,), kv, dtype=torch.int32, device="cuda")
    qo_indptr = torch.arange(0, bs + 1, dtype=torch.int32, device="cuda")
    ckv = torch.randn(npages, page, 512, dtype=torch.bfloat16, device="cuda")
    kpe = torch.randn(npages, page, 64, dtype=torch.bfloat16, device="cuda")
    q_nope = torch.randn(bs, H, 512, dtype=torch.bfloat16, device="cuda")
    q_pe = torch.randn(bs, H, 64, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(2048 * 1024 * 1024, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    sm = 1.0 / math.sqrt(576)
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, H, 512, 64, page, True, sm, torch.bfloat16)
    out = torch.empty(bs, H, 512, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: w.run(q_nope, q_pe, ckv, kpe, out=out))
    fl = 2 * bs * H * kv * (576 + 512)
    bytes_kv = bs * kv * 576 * 2
    print(f"mla decode bs={bs} kv={kv} H={H}: {t*1e6:.1f} us  {fl/t/1e12:.1f} TFLOPS  {bytes_kv/t/1e12:.2f} TB/s")


def bench_moe(T=4096, H=4096, inter=14336, E=8, k=2):
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    torch.manual_seed(0)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 16
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    t = timeit(lambda: fused_moe(x, w13, w2, weights, ids), iters=10, warmup=3)
    fl = T * k * 3 * H * inter * 2
    print(f"fused_moe mixtral T={T}: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TFLOPS  {T/t/1e6:.3f} M tok/s")




def bench_moe_fp8(T=4096, H=4096, inter=14336, E=8, k=2):
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import per_block_quant_fp8
    torch.manual_seed(0)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 16
    w13_q, w13_s = per_block_quant_fp8(w13)
    w2_q, w2_s = per_block_quant_fp8(w2)
    del w13, w2
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    t = timeit(lambda: fused_moe(x, w13_q, w2_q, weights, ids, w13_scale=w13_s, w2_scale=w2_s), iters=10, warmup=3)
    fl = T * k * 3 * H * inter * 2
    print(f"fused_moe fp8 mixtral T={T}: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TFLOPS  {T/t/1e6:.3f} M tok/s")


def bench_fp8_gemm(M=4096, N=4096, K=4096):
    from flashinfer_amd.fp8_quantization import (gemm_fp8_nt_groupwise, per_block_quant_fp8, per_token_group_quant_fp8)
    torch.manual_seed(0)
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    a_q, a_s = per_token_group_quant_fp8(a, transpose_scale=True)
    b_q, b_s = per_block_quant_fp8(b)
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: gemm_fp8_nt_groupwise(a_q, b_q, a_s, b_s, out=out), iters=10)
    print(f"fp8 groupwise gemm {M}x{N}x{K}: {t*1e3:.2f} ms  {2*M*N*K/t/1e12:.0f} TFLOPS")


def bench_prefill_ragged_192(bs=16, s=1024, Hq=128, Hkv=128, causal=True):
    """DeepSeek MHA ragged prefill, hd_qk=192/hd_vo=128 (BASELINE.md ragged
    row; B200 fa2 = 213.7 TF)."""
    torch.manual_seed(0)
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    q = torch.randn(bs * s, Hq, 192, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(bs * s, Hkv, 192, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(bs * s, Hkv, 128, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, qo_indptr.clone(), Hq, Hkv, 192, head_dim_vo=128,
           causal=causal, q_data_type=torch.bfloat16)
    out = torch.empty(bs * s, Hq, 128, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: w.run(q, k, v, out=out))
    fl = bs * Hq * s * s * 2 * (192 + 128)
    if causal:
        fl /= 2
    print(f"prefill ragged 192/128 bs={bs} s={s} H={Hq} causal={causal}: "
          f"{t*1e6:.1f} us  {fl/t/1e12:.1f} TFLOPS")


def bench_prefill_splitkv(qo=16, kv=65536, Hq=32, Hkv=8, D=128):
    """bs=1 short-q/long-kv prefill: split-KV must fill the chip."""
    torch.manual_seed(0)
    qo_indptr = torch.tensor([0, qo], dtype=torch.int32, device="cuda")
    kv_indptr = torch.tensor([0, kv], dtype=torch.int32, device="cuda")
    q = torch.randn(qo, Hq, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(kv, Hkv, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchPrefillWithRaggedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, Hq, Hkv, D, causal=False,
           q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    t = timeit(lambda: w.run(q, k, v, out=out))
    bytes_kv = kv * Hkv * D * 2 * 2
    print(f"prefill splitkv qo={qo} kv={kv} (split={w._split}): {t*1e6:.1f} us  "
          f"{bytes_kv/t/1e12:.2f} TB/s")


def bench_decode_modes(bs=16, kv=1024, Hq=64, Hkv=8, D=128, page=16):
    """A/B the three decode shapes (fused / split-vector / tensor-core) at one
    config — the BASELINE small-batch decode latency row."""
    torch.manual_seed(0)
    pages_per = (kv + page - 1) // page
    kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per,
                             dtype=torch.int32, device="cuda")
    npages = bs * pages_per
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    last_page = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32,
                           device="cuda")
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device="cuda")
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(512 << 20, dtype=torch.uint8, device="cuda")
    bytes_kv = bs * kv * Hkv * D * 2 * 2
    out = torch.empty_like(q)
    for mode, kwargs in (("auto", {}), ("vector", dict(use_tensor_cores=False)),
                         ("tc", dict(use_tensor_cores=True))):
        from flashinfer_amd import decode as _dec
        saved = _dec._FUSED_MAX_KV
        if mode == "vector":
            _dec._FUSED_MAX_KV = 0  # force the split vector path
        w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD", **kwargs)
        w.plan(kv_indptr, kv_indices, last_page, Hq, Hkv, D, page,
               q_data_type=torch.bfloat16)
        _dec._FUSED_MAX_KV = saved
        t = timeit(lambda: w.run(q, (k_cache, v_cache), out=out))
        tag = ("mfma" if getattr(w, "_fused_mfma", False) else
               "fused" if getattr(w, "_fused", False) else
               "tc" if getattr(w, "_tc", False) else "vector")
        print(f"decode_modes bs={bs} kv={kv} GQA{Hq}/{Hkv} [{mode}->{tag}]: "
              f"{t*1e6:.1f} us  {bytes_kv/t/1e12:.2f} TB/s")



def bench_holistic(n_prefill=8, n_decode=120, s_prefill=1024, kv_decode=1024,
                   Hq=64, Hkv=8, D=128, page=16):
    """Persistent holistic BatchAttention vs the two-wrapper composition on a
    mixed prefill+decode batch (VERDICT #2 acceptance measurement)."""
    torch.manual_seed(0)
    qo_lens = [s_prefill] * n_prefill + [1] * n_decode
    kv_lens = [s_prefill] * n_prefill + [kv_decode] * n_decode
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
    out = torch.empty_like(q)
    # holistic single launch
    w = fi.BatchAttention("NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, Hq, Hkv, D, D, page,
           causal=True, q_data_type=torch.bfloat16)
    t_h = timeit(lambda: w.run(q, (kc, vc), out=out, return_lse=False))
    # two-wrapper composition (prefill wrapper for prefills + decode wrapper)
    lp = ((kv_len_arr.to(torch.int64) - 1) % page + 1).to(torch.int32)
    wp = fi.BatchPrefillWithPagedKVCacheWrapper(
        torch.empty(256 << 20, dtype=torch.uint8, device="cuda"), "NHD")
    wp.plan(qo_indptr[: n_prefill + 1], kv_indptr[: n_prefill + 1], kv_indices,
            lp[:n_prefill], Hq, Hkv, D, page, causal=True,
            q_data_type=torch.bfloat16)
    wd = fi.BatchDecodeWithPagedKVCacheWrapper(
        torch.empty(256 << 20, dtype=torch.uint8, device="cuda"), "NHD")
    dec_indptr = (kv_indptr[n_prefill:] - kv_indptr[n_prefill]).contiguous()
    wd.plan(dec_indptr, kv_indices[int(kv_indptr[n_prefill]):], lp[n_prefill:],
            Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    q_pf = q[: qo_indptr[n_prefill]]
    q_dec = q[qo_indptr[n_prefill]:]
    o_pf = out[: qo_indptr[n_prefill]]
    o_dec = out[qo_indptr[n_prefill]:]
    def two():
        wp.run(q_pf, (kc, vc), out=o_pf)
        wd.run(q_dec, (kc, vc), out=o_dec)
    t_2 = timeit(two)
    # concurrent: prefill and decode kernels on separate streams
    s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()
    def conc():
        ev = torch.cuda.Event()
        with torch.cuda.stream(s1):
            wp.run(q_pf, (kc, vc), out=o_pf)
        with torch.cuda.stream(s2):
            wd.run(q_dec, (kc, vc), out=o_dec)
        torch.cuda.current_stream().wait_stream(s1)
        torch.cuda.current_stream().wait_stream(s2)
    t_c = timeit(conc)
    fl = sum(2 * 2 * Hq * D * q_ * k_ / (2 if q_ > 1 else 1)
             for q_, k_ in zip(qo_lens, kv_lens))
    print(f"holistic {n_prefill}pf/{n_decode}dec: one-launch {t_h*1e6:.0f} us "
          f"({fl/t_h/1e12:.1f} TF)  two-wrapper {t_2*1e6:.0f} us  "
          f"two-stream {t_c*1e6:.0f} us  speedup {t_2/t_h:.2f}x")



def bench_mla_prefill(bs=16, s=1024, H=128, page=32):
    """Full chunked MLA prefill (qo = kv = s) — reference mla.cuh:976 role."""
    import math
    torch.manual_seed(0)
    pages_per = (s + page - 1) // page
    kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32, device="cuda")
    npages = bs * pages_per
    kv_indices = torch.randperm(npages, dtype=torch.int32, device="cuda")
    kv_len_arr = torch.full((bs,), s, dtype=torch.int32, device="cuda")
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device="cuda")
    ckv = torch.randn(npages, page, 512, dtype=torch.bfloat16, device="cuda") / 4
    kpe = torch.randn(npages, page, 64, dtype=torch.bfloat16, device="cuda") / 4
    q_nope = torch.randn(bs * s, H, 512, dtype=torch.bfloat16, device="cuda") / 4
    q_pe = torch.randn(bs * s, H, 64, dtype=torch.bfloat16, device="cuda") / 4
    ws = torch.empty(4096 << 20, dtype=torch.uint8, device="cuda")
    w = fi.BatchMLAPagedAttentionWrapper(ws)
    sm = 1.0 / math.sqrt(576)
    w.plan(qo_indptr, kv_indptr, kv_indices, kv_len_arr, H, 512, 64, page, True, sm, torch.bfloat16)
    out = torch.empty(bs * s, H, 512, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: w.run(q_nope, q_pe, ckv, kpe, out=out), iters=10, warmup=3)
    fl = bs * H * s * s / 2 * 2 * (576 + 512)
    print(f"mla prefill bs={bs} s={s} H={H}: {t*1e3:.2f} ms  {fl/t/1e12:.1f} TFLOPS")


def bench_moe_mx(T=4096, H=4096, inter=14336, E=8, k=2):
    """MX-fp8 MoE: e8m0 hardware scales in the f8f6f4 MFMA (no rescale VALU)."""
    from flashinfer_amd.fused_moe import fused_moe, moe_topk_softmax
    from flashinfer_amd.fp8_quantization import per_block_quant_mxfp8
    torch.manual_seed(0)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") / 4
    w13 = torch.randn(E, 2 * inter, H, dtype=torch.bfloat16, device="cuda") / 16
    w2 = torch.randn(E, H, inter, dtype=torch.bfloat16, device="cuda") / 16
    w13_q, w13_s = per_block_quant_mxfp8(w13)
    w2_q, w2_s = per_block_quant_mxfp8(w2)
    del w13, w2
    logits = torch.randn(T, E, device="cuda")
    weights, ids = moe_topk_softmax(logits, k)
    t = timeit(lambda: fused_moe(x, w13_q, w2_q, weights, ids, w13_scale=w13_s, w2_scale=w2_s), iters=10, warmup=3)
    fl = T * k * 3 * H * inter * 2
    print(f"fused_moe MX-fp8 mixtral T={T}: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TFLOPS  {T/t/1e6:.3f} M tok/s")


def _clock_warm(seconds=0.7):
    """DVFS reaches steady clocks only after ~0.5-1 s of sustained load;
    without this the first benches in a sweep read ~10% low."""
    import time as _t

    a = torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda")
    t0 = _t.perf_counter()
    while _t.perf_counter() - t0 < seconds:
        a = a @ a * 1e-3
    torch.cuda.synchronize()


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    _clock_warm()
    if which in ("all", "prefill"):
        bench_prefill()
        bench_prefill(bs=1, s=8192)
        bench_prefill(bs=16, s=1024, causal=False)
        bench_prefill_ragged_192()
        bench_prefill_splitkv()
    if which in ("all", "decode"):
        bench_decode()
        bench_decode(bs=16, kv=1024)
        bench_decode(bs=128, kv=4096)
        # GQA-8 (Llama-70B class): auto tensor-core route
        bench_decode(bs=256, kv=8192, Hq=64, Hkv=8)
        bench_decode(bs=16, kv=1024, Hq=64, Hkv=8)
    if which in ("all", "modes", "decode"):
        bench_decode_modes()                       # BASELINE small-batch row
        bench_decode_modes(bs=16, kv=1024, Hq=32, Hkv=8)
        bench_decode_modes(bs=64, kv=1024, Hq=64, Hkv=8)
        bench_decode_modes(bs=256, kv=2048, Hq=64, Hkv=8)
        bench_decode_modes(bs=32, kv=512, Hq=32, Hkv=8)
    if which in ("all", "gemm"):
        bench_gemm(4096)
        bench_gemm(8192)
    if which in ("all", "mla"):
        bench_mla()
        bench_mla(bs=64, kv=4096)
        bench_mla_prefill(bs=1)
        bench_mla_prefill(bs=16)
    if which in ("all", "holistic"):
        bench_holistic()
        bench_holistic(n_prefill=2, n_decode=200, kv_decode=2048)
        bench_holistic(n_prefill=16, n_decode=16)
    if which in ("all", "moe"):
        bench_moe()
        bench_moe_fp8()
        bench_moe_mx()
    if which in ("all", "fp8"):
        bench_fp8_gemm()
