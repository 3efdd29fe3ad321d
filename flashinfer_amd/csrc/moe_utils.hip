// MoE data-movement kernels: token-copy gather (expand/permute) and the
// finalize scatter-reduce (reference parity: nv_internal moe_kernels
// expand/permute + finalize stages, csrc/fused_moe). Pure bandwidth ops:
// 16-byte lane vectors, grid-stride.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

// dst[r, :] = src[row_map[r], :]
template <typename T, int VEC>
__global__ void gather_rows_kernel(const T* __restrict__ src, T* __restrict__ dst,
                                   const int32_t* __restrict__ row_map, int64_t rows,
                                   int cols) {
  int chunks = cols / VEC;
  int64_t total = rows * chunks;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(idx % chunks) * VEC;
    int64_t r = idx / chunks;
    vec_t<T, VEC> v;
    v.load(src + (int64_t)row_map[r] * cols + c);
    v.store(dst + r * cols + c);
  }
}

// out[t, :] = sum_j w[t, j] * h[pos[t, j], :]
template <typename T, int VEC>
__global__ void moe_finalize_kernel(const T* __restrict__ h, T* __restrict__ out,
                                    const int32_t* __restrict__ pos,
                                    const float* __restrict__ w, int64_t tokens,
                                    int topk, int cols) {
  int chunks = cols / VEC;
  int64_t total = tokens * chunks;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(idx % chunks) * VEC;
    int64_t t = idx / chunks;
    float acc[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
    for (int k = 0; k < topk; ++k) {
      float wk = w[t * topk + k];
      vec_t<T, VEC> v;
      v.load(h + (int64_t)pos[t * topk + k] * cols + c);
#pragma unroll
      for (int j = 0; j < VEC; ++j) acc[j] += wk * v.get(j);
    }
    vec_t<T, VEC> o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) o.set(j, acc[j]);
    o.store(out + t * cols + c);
  }
}

}  // namespace fi

extern "C" hipError_t fi_gather_rows(int dtype, const void* src, void* dst,
                                     const int32_t* row_map, int64_t rows, int cols,
                                     hipStream_t stream) {
  if (rows == 0) return hipSuccess;
  int vec = cols % 8 == 0 ? 8 : 1;
  int64_t total = rows * (cols / vec);
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
#define LG(T, V)                                                               \
  hipLaunchKernelGGL((fi::gather_rows_kernel<T, V>), dim3(grid), dim3(256), 0, \
                     stream, (const T*)src, (T*)dst, row_map, rows, cols)
  switch (dtype * 2 + (vec == 8)) {
    case 1: LG(fi::bf16, 8); break;
    case 0: LG(fi::bf16, 1); break;
    case 3: LG(fi::fp16, 8); break;
    case 2: LG(fi::fp16, 1); break;
    case 5: LG(float, 8); break;
    case 4: LG(float, 1); break;
    default: return hipErrorInvalidValue;
  }
#undef LG
  return hipGetLastError();
}

extern "C" hipError_t fi_moe_finalize(int dtype, const void* h, void* out,
                                      const int32_t* pos, const float* w,
                                      int64_t tokens, int topk, int cols,
                                      hipStream_t stream) {
  if (tokens == 0) return hipSuccess;
  int vec = cols % 8 == 0 ? 8 : 1;
  int64_t total = tokens * (cols / vec);
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
#define LF(T, V)                                                                \
  hipLaunchKernelGGL((fi::moe_finalize_kernel<T, V>), dim3(grid), dim3(256), 0, \
                     stream, (const T*)h, (T*)out, pos, w, tokens, topk, cols)
  switch (dtype * 2 + (vec == 8)) {
    case 1: LF(fi::bf16, 8); break;
    case 0: LF(fi::bf16, 1); break;
    case 3: LF(fi::fp16, 8); break;
    case 2: LF(fi::fp16, 1); break;
    case 5: LF(float, 8); break;
    case 4: LF(float, 1); break;
    default: return hipErrorInvalidValue;
  }
#undef LF
  return hipGetLastError();
}
