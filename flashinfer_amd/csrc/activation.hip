// Fused gated activations: silu_and_mul / gelu_and_mul / gelu_tanh_and_mul.
// Parity with reference include/flashinfer/activation.cuh (act_and_mul_kernel:29).
// Input [tokens, 2*d] (gate | up), output [tokens, d]: out = act(gate) * up.
// Memory-bound: bf16x8 lane loads, grid-stride.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

struct SiluOp {
  __device__ static float apply(float x) { return x / (1.f + __builtin_expf(-x)); }
};
struct GeluOp {  // exact erf-based
  __device__ static float apply(float x) { return 0.5f * x * (1.f + erff(x * 0.70710678118654752f)); }
};
struct GeluTanhOp {
  __device__ static float apply(float x) {
    float x3 = x * x * x;
    return 0.5f * x * (1.f + tanhf(0.79788456080286536f * (x + 0.044715f * x3)));
  }
};

template <typename T, typename Op, int VEC>
__global__ void act_and_mul_kernel(const T* __restrict__ in, T* __restrict__ out,
                                   int64_t tokens, int d) {
  int64_t total = tokens * (int64_t)(d / VEC);
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t token = idx / (d / VEC);
    int off = (int)(idx % (d / VEC)) * VEC;
    vec_t<T, VEC> g, u, o;
    g.load(in + token * 2 * d + off);
    u.load(in + token * 2 * d + d + off);
#pragma unroll
    for (int j = 0; j < VEC; ++j) o.set(j, Op::apply(g.get(j)) * u.get(j));
    o.store(out + token * d + off);
  }
}

template <typename T>
hipError_t act_launch(int which, const void* in, void* out, int64_t tokens, int d,
                      hipStream_t stream) {
  int vec = (d % 8 == 0) ? 8 : 1;
  int64_t total = tokens * (d / vec);
  int grid = (int)((total + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid == 0) grid = 1;
  dim3 g(grid), blk(256);
#define LAUNCH_A(OP)                                                                      \
  do {                                                                                    \
    if (vec == 8)                                                                         \
      hipLaunchKernelGGL((act_and_mul_kernel<T, OP, 8>), g, blk, 0, stream, (const T*)in, \
                         (T*)out, tokens, d);                                             \
    else                                                                                  \
      hipLaunchKernelGGL((act_and_mul_kernel<T, OP, 1>), g, blk, 0, stream, (const T*)in, \
                         (T*)out, tokens, d);                                             \
  } while (0)
  switch (which) {
    case 0: LAUNCH_A(SiluOp); break;
    case 1: LAUNCH_A(GeluOp); break;
    case 2: LAUNCH_A(GeluTanhOp); break;
  }
#undef LAUNCH_A
  return hipGetLastError();
}

}  // namespace fi

// which: 0 silu, 1 gelu, 2 gelu_tanh; dtype: 0 bf16, 1 fp16, 2 fp32
extern "C" hipError_t fi_act_and_mul(int which, int dtype, const void* in, void* out,
                                     int64_t tokens, int d, hipStream_t stream) {
  switch (dtype) {
    case 0: return fi::act_launch<fi::bf16>(which, in, out, tokens, d, stream);
    case 1: return fi::act_launch<fi::fp16>(which, in, out, tokens, d, stream);
    case 2: return fi::act_launch<float>(which, in, out, tokens, d, stream);
  }
  return hipErrorInvalidValue;
}
