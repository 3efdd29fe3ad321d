"""Driver benchmark contract: the reference's headline metric on MI355X.

Metric (BASELINE.json): "paged prefill TFLOPS + batch-decode tok/s,
Llama-3-8B GQA head_dim=128".

Headline value = batch-decode tokens/s (whole-job aggregate over N GPUs),
measured over EXACTLY --steps timed decode iterations of the long-context
config (batch=256, kv_len=32768, page=16, GQA 32q/8kv, head_dim=128, bf16,
synthetic data). Paged prefill TFLOPS (bs=16, s=1024, causal) is measured in
a separate untimed-region phase and reported as an auxiliary key.

Run:  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches via torch.distributed.run (one rank per GPU over
RCCL); weak scaling — each rank runs the fixed per-GPU workload.
"""
import argparse
import json
import os
import sys

import torch


def build_decode(dev, bs=256, kv=32768, Hq=32, Hkv=8, D=128, page=16):
    import flashinfer_amd as fi

    pages_per = (kv + page - 1) // page
    indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32,
                          device=dev)
    npages = bs * pages_per
    indices = torch.randperm(npages, dtype=torch.int32, device=dev)
    last_page = torch.full((bs,), (kv - 1) % page + 1, dtype=torch.int32, device=dev)
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev)
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(bs, Hq, D, dtype=torch.bfloat16, device=dev)
    ws = torch.empty(512 * 1024 * 1024, dtype=torch.uint8, device=dev)
    w = fi.BatchDecodeWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(indptr, indices, last_page, Hq, Hkv, D, page, q_data_type=torch.bfloat16)
    out = torch.empty_like(q)
    return lambda: w.run(q, (k_cache, v_cache), out=out)


def build_prefill(dev, bs=16, s=1024, Hq=32, Hkv=8, D=128, page=16):
    import flashinfer_amd as fi

    pages_per = (s + page - 1) // page
    qo_indptr = torch.arange(0, (bs + 1) * s, s, dtype=torch.int32, device=dev)
    kv_indptr = torch.arange(0, (bs + 1) * pages_per, pages_per, dtype=torch.int32,
                             device=dev)
    npages = bs * pages_per
    kv_indices = torch.randperm(npages, dtype=torch.int32, device=dev)
    last_page = torch.full((bs,), (s - 1) % page + 1, dtype=torch.int32, device=dev)
    k_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev)
    v_cache = torch.randn(npages, page, Hkv, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(bs * s, Hq, D, dtype=torch.bfloat16, device=dev)
    ws = torch.empty(64 * 1024 * 1024, dtype=torch.uint8, device=dev)
    w = fi.BatchPrefillWithPagedKVCacheWrapper(ws, "NHD")
    w.plan(qo_indptr, kv_indptr, kv_indices, last_page, Hq, Hkv, D, page, causal=True)
    out = torch.empty_like(q)
    # causal flops: 2 (QK) + 2 (PV) MACs over the lower triangle
    flops = bs * Hq * D * 4 * (s * (s + 1) / 2)
    return (lambda: w.run(q, (k_cache, v_cache), out=out)), flops


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world_size)

    dist = None
    if world_size > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group("nccl")
        torch.cuda.set_device(local_rank)
    dev = f"cuda:{local_rank}"
    torch.manual_seed(1234 + rank)

    bs, kv = 256, 32768
    decode_step = build_decode(dev, bs=bs, kv=kv)
    prefill_step, prefill_flops = build_prefill(dev)

    # aux phase: prefill TFLOPS (not part of the official timed region)
    for _ in range(3):
        prefill_step()
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(10):
        prefill_step()
    e1.record()
    torch.cuda.synchronize()
    prefill_t = e0.elapsed_time(e1) / 10 * 1e-3
    prefill_tflops = prefill_flops / prefill_t / 1e12

    # multi-GPU aux phase (untimed): custom-AR latency (one-shot/two-shot
    # over hipIpc + xGMI) and MoE all-to-all (RCCL alltoallv) — the VERDICT
    # r01 pre-staged comm measurements; failure-guarded so a comm issue can
    # never sink the headline bench.
    comm_aux = {}
    if dist and world_size > 1:
        try:
            from flashinfer_amd.comm.custom_ar import CustomAllReduce

            ar = CustomAllReduce(max_bytes=16 << 20)
            import time as _t

            for label, numel in (("8KiB", 4096), ("128KiB", 65536),
                                 ("2MiB", 1 << 20), ("16MiB", 8 << 20)):
                x = torch.randn(numel, dtype=torch.bfloat16, device=dev)
                for strat in ("one_shot", "two_shot"):
                    for _ in range(5):
                        ar.all_reduce(x, strategy=strat)
                    torch.cuda.synchronize()
                    dist.barrier()
                    t0_ = _t.perf_counter()
                    for _ in range(20):
                        ar.all_reduce(x, strategy=strat)
                    torch.cuda.synchronize()
                    dt = (_t.perf_counter() - t0_) / 20
                    comm_aux[f"ar_{strat}_{label}_us"] = round(dt * 1e6, 1)
            # RCCL comparison at the same sizes
            for label, numel in (("128KiB", 65536), ("16MiB", 8 << 20)):
                x = torch.randn(numel, dtype=torch.bfloat16, device=dev)
                for _ in range(5):
                    dist.all_reduce(x)
                torch.cuda.synchronize()
                dist.barrier()
                t0_ = _t.perf_counter()
                for _ in range(20):
                    dist.all_reduce(x)
                torch.cuda.synchronize()
                comm_aux[f"ar_rccl_{label}_us"] = round(
                    (_t.perf_counter() - t0_) / 20 * 1e6, 1)
            ar.close()
            # MoE a2a (alltoallv-shaped single-tensor exchange, 1 MiB/rank)
            xa = torch.randn(world_size * 65536, dtype=torch.bfloat16,
                             device=dev)
            ya = torch.empty_like(xa)
            for _ in range(5):
                dist.all_to_all_single(ya, xa)
            torch.cuda.synchronize()
            dist.barrier()
            t0_ = _t.perf_counter()
            for _ in range(20):
                dist.all_to_all_single(ya, xa)
            torch.cuda.synchronize()
            comm_aux["moe_a2a_128KiB_per_rank_us"] = round(
                (_t.perf_counter() - t0_) / 20 * 1e6, 1)
        except Exception as e:  # degrade, never sink the headline
            comm_aux["error"] = repr(e)[:200]

    # clock ramp: MI355X DVFS takes ~0.5-1 s of sustained load to reach
    # steady clocks (measured: the first bench iterations read ~10% low —
    # profiles/README r02 "warm-clock confirmations"). Spin a dummy GEMM
    # for ~0.7 s BEFORE the untimed warmup so the W warmup steps and the
    # K timed steps both run at steady state. Untimed, outside the
    # measured region, and touches none of the benchmarked tensors.
    import time

    _wa = torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda")
    _wt = time.perf_counter()
    while time.perf_counter() - _wt < 0.7:
        _wa = _wa @ _wa * 1e-3
    torch.cuda.synchronize()
    del _wa

    # warmup
    for _ in range(args.warmup):
        decode_step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
        torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        decode_step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    if dist:
        t = torch.tensor([elapsed], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tok_s = bs * args.steps * n_gpus / elapsed
    ms_per_step = elapsed / args.steps * 1e3
    decode_tb_s = bs * kv * 8 * 128 * 2 * 2 * (1.0 / (elapsed / args.steps)) / 1e12

    if rank == 0:
        print(json.dumps({
            "metric": "paged prefill TFLOPS + batch-decode tok/s, Llama-3-8B GQA head_dim=128",
            "value": tok_s,
            "unit": "decode_tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic (random-init weights/cache)",
            "prefill_tflops_per_gpu": prefill_tflops,
            "decode_tb_per_s_per_gpu": decode_tb_s,
            **({"comm_aux": comm_aux} if comm_aux else {}),
            "config": {
                "model": "Llama-3-8B attention (GQA 32q/8kv, head_dim=128)",
                "decode_batch": bs, "decode_kv_len": kv, "page_size": 16,
                "prefill_batch": 16, "prefill_seq_len": 1024,
                "parallelism": f"dp{n_gpus}",
            },
        }))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
