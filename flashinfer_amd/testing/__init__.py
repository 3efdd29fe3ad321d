"""Benchmark / testing utilities (reference parity: flashinfer/testing/
utils.py bench_gpu_time:774 family and the attention flops/bytes
calculators:456-750)."""
from .utils import (
    attention_tb_per_sec_with_actual_seq_lens,
    attention_tflops_per_sec_with_actual_seq_lens,
    bench_gpu_time,
    bench_gpu_time_with_cuda_event,
    bench_gpu_time_with_cudagraph,
)
