"""GPU micro-benchmark helpers for MI355X (reference parity:
flashinfer/testing/utils.py — bench_gpu_time:774,
bench_gpu_time_with_cuda_event:,
bench_gpu_time_with_cudagraph:, attention calculators:456-750). Timing uses
HIP events; an L2-flush rotation buffer defeats the per-XCD L2 caches
between iterations the same way the reference's rotation buffers defeat the
H100 L2."""
from __future__ import annotations

from typing import Callable, List, Optional

import torch


def _l2_flush_buffer(device) -> torch.Tensor:
    # 8 XCDs x 4 MB L2 each -> 128 MB wipes every slice comfortably
    return torch.empty(128 * 1024 * 1024, dtype=torch.uint8, device=device)


def bench_gpu_time_with_cuda_event(
    fn: Callable[[], None],
    dry_run_iters: int = 10,
    repeat_iters: int = 50,
    l2_flush: bool = True,
    l2_flush_device: str = "cuda",
    sleep_after_run: bool = False,
) -> List[float]:
    r"""Median-friendly per-iteration times (ms) via HIP events, optionally
    flushing L2 between iterations."""
    flush = _l2_flush_buffer(l2_flush_device) if l2_flush else None
    for _ in range(dry_run_iters):
        fn()
    torch.cuda.synchronize()
    times = []
    for _ in range(repeat_iters):
        if flush is not None:
            flush.zero_()
        start = torch.cuda.Event(enable_timing=True)
        stop = torch.cuda.Event(enable_timing=True)
        start.record()
        fn()
        stop.record()
        torch.cuda.synchronize()
        times.append(start.elapsed_time(stop))
    return times


bench_gpu_time = bench_gpu_time_with_cuda_event


def bench_gpu_time_with_cudagraph(
    fn: Callable[[], None],
    dry_run_iters: int = 3,
    repeat_iters: int = 20,
    num_iters_within_graph: int = 10,
    l2_flush: bool = True,
    l2_flush_device: str = "cuda",
    sleep_after_run: bool = False,
) -> List[float]:
    r"""Times ``num_iters_within_graph`` launches captured in one hipGraph —
    isolates kernel time from launch overhead (reference
    bench_gpu_time_with_cudagraph role)."""
    g = torch.cuda.CUDAGraph()
    fn()
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        for _ in range(num_iters_within_graph):
            fn()
    times = bench_gpu_time_with_cuda_event(
        g.replay, dry_run_iters, repeat_iters, l2_flush, l2_flush_device)
    return [t / num_iters_within_graph for t in times]


def attention_flops(batch_size, qo_len, kv_len, head_dim_qk, head_dim_vo,
                    num_qo_heads, causal: bool = False) -> float:
    r"""Total FLOPs of one attention call (2*QK^T + 2*PV per head-row)."""
    if causal:
        valid = kv_len * qo_len - qo_len * (qo_len - 1) / 2
    else:
        valid = kv_len * qo_len
    per_head = 2 * valid * (head_dim_qk + head_dim_vo)
    return float(batch_size * num_qo_heads * per_head)


def attention_tflops_per_sec_with_actual_seq_lens(
    actual_seq_lens_q, actual_seq_lens_kv, head_dim_qk, head_dim_vo,
    num_qo_heads, causal, time_ms: float,
) -> float:
    f = sum(
        attention_flops(1, int(lq), int(lk), head_dim_qk, head_dim_vo,
                        num_qo_heads, causal)
        for lq, lk in zip(actual_seq_lens_q.flatten().tolist(),
                          actual_seq_lens_kv.flatten().tolist())
    )
    return f / (time_ms * 1e-3) / 1e12


def attention_tb_per_sec_with_actual_seq_lens(
    actual_seq_lens_q, actual_seq_lens_kv, head_dim_qk, head_dim_vo,
    num_qo_heads, num_kv_heads, time_ms: float,
    q_dtype_bytes: int = 2, kv_dtype_bytes: int = 2, o_dtype_bytes: int = 2,
) -> float:
    r"""Achieved memory bandwidth assuming Q/O read+written once and K/V
    streamed once (the decode/append bound on HBM3E)."""
    lq = actual_seq_lens_q.flatten().to(torch.float64)
    lk = actual_seq_lens_kv.flatten().to(torch.float64)
    q_bytes = float(lq.sum()) * num_qo_heads * head_dim_qk * q_dtype_bytes
    o_bytes = float(lq.sum()) * num_qo_heads * head_dim_vo * o_dtype_bytes
    kv_bytes = float(lk.sum()) * num_kv_heads * (
        head_dim_qk + head_dim_vo) * kv_dtype_bytes
    return (q_bytes + o_bytes + kv_bytes) / (time_ms * 1e-3) / 1e12
