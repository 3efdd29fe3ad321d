"""JIT layer: build on CPU (hipcc cross-compiles gfx950), execute on GPU."""
import ctypes

import pytest
import torch

SRC = r"""
#include "fi/common.hpp"
#include "fi/vec.hpp"

// a custom fused op built from the library's device headers
__global__ void scale_bias_kernel(const fi::bf16* x, fi::bf16* y, float a, float b,
                                  int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    y[i] = fi::from_f32<fi::bf16>(fi::to_f32<fi::bf16>(x[i]) * a + b);
  }
}

extern "C" int custom_scale_bias(const void* x, void* y, float a, float b, int64_t n,
                                 void* stream) {
  hipLaunchKernelGGL(scale_bias_kernel, dim3(256), dim3(256), 0, (hipStream_t)stream,
                     (const fi::bf16*)x, (fi::bf16*)y, a, b, n);
  return (int)hipGetLastError();
}
"""


def _spec(tmp_path_factory=None):
    from flashinfer_amd.jit import gen_jit_spec

    return gen_jit_spec("test_scale_bias", {"scale_bias.hip": SRC})


def test_jit_build_cpu():
    spec = _spec()
    spec.build()
    assert spec.so_path.exists()
    # cache hit: second build_and_load is instant and loads
    lib = spec.build_and_load()
    assert hasattr(lib, "custom_scale_bias")


@pytest.mark.gpu
def test_jit_custom_op_runs():
    spec = _spec()
    lib = spec.build_and_load()
    x = torch.randn(10000, dtype=torch.bfloat16, device="cuda")
    y = torch.empty_like(x)
    stream = torch.cuda.current_stream().cuda_stream
    rc = lib.custom_scale_bias(
        ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(y.data_ptr()),
        ctypes.c_float(2.0), ctypes.c_float(1.0), ctypes.c_int64(x.numel()),
        ctypes.c_void_p(stream),
    )
    assert rc == 0
    torch.cuda.synchronize()
    torch.testing.assert_close(y.float(), x.float() * 2 + 1, atol=1e-2, rtol=1e-2)
