// Top-k selection + bitmask packing for gfx950.
// Parity: reference include/flashinfer/topk.cuh (RadixTopKKernel_Unified:1270
// — ours uses the float-bit-monotonic threshold search + compaction, same
// sorting-free contract: unordered top-k) and quantization.cuh
// (PackBitsKernel:40, SegmentPackBitsKernel:66).
#include "fi/common.hpp"

namespace fi {

constexpr int TB = 512;
constexpr int TW = TB / kWaveSize;

__device__ __forceinline__ float tk_block_sum(float x, float* smem) {
  x = wave_reduce_sum<kWaveSize>(x);
  int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
  if (l == 0) smem[w] = x;
  __syncthreads();
  float r = 0.f;
  if (threadIdx.x == 0) {
#pragma unroll
    for (int i = 0; i < TW; ++i) r += smem[i];
    smem[TW] = r;
  }
  __syncthreads();
  r = smem[TW];
  __syncthreads();
  return r;
}

__device__ __forceinline__ uint32_t tk_f2u(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float tk_u2f(uint32_t u) {
  return __uint_as_float((u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u);
}

// one block per row; unordered top-k (values + indices)
__global__ void topk_kernel(const float* __restrict__ x, float* __restrict__ out_v,
                            int32_t* __restrict__ out_i, int rows, int d, int k,
                            int64_t stride) {
  __shared__ float smem[TW + 1];
  __shared__ int s_cnt;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const float* xr = x + (int64_t)row * stride;
    // threshold tau: largest value with |{v >= tau}| >= k
    uint32_t lo = 0, hi = 0xFFFFFFFFu;
    for (int it = 0; it < 24 && lo < hi; ++it) {
      uint32_t mid = lo + ((hi - lo) >> 1);
      float tau = tk_u2f(mid);
      float cnt = 0.f;
      for (int i = threadIdx.x; i < d; i += TB)
        if (xr[i] >= tau) cnt += 1.f;
      float g = tk_block_sum(cnt, smem);
      if (g >= (float)k) lo = mid;
      else hi = mid - 1;
    }
    float tau = tk_u2f(lo);
    // compact: strictly-greater first, then fill ties
    if (threadIdx.x == 0) s_cnt = 0;
    __syncthreads();
    for (int i = threadIdx.x; i < d; i += TB) {
      float v = xr[i];
      if (v > tau) {
        int pos = atomicAdd(&s_cnt, 1);
        if (pos < k) {
          out_v[(int64_t)row * k + pos] = v;
          out_i[(int64_t)row * k + pos] = i;
        }
      }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < d; i += TB) {
      float v = xr[i];
      if (v == tau) {
        int pos = atomicAdd(&s_cnt, 1);
        if (pos < k) {
          out_v[(int64_t)row * k + pos] = v;
          out_i[(int64_t)row * k + pos] = i;
        }
      }
    }
    __syncthreads();
  }
}

// bitorder "little": bit j of output byte b = x[b*8 + j]
__global__ void packbits_kernel(const uint8_t* __restrict__ x, uint8_t* __restrict__ y,
                                int64_t n) {
  int64_t nbytes = (n + 7) / 8;
  for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; b < nbytes;
       b += (int64_t)gridDim.x * blockDim.x) {
    uint8_t v = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int64_t i = b * 8 + j;
      if (i < n && x[i]) v |= (1u << j);
    }
    y[b] = v;
  }
}

__global__ void segment_packbits_kernel(const uint8_t* __restrict__ x,
                                        uint8_t* __restrict__ y,
                                        const int32_t* __restrict__ x_indptr,
                                        const int32_t* __restrict__ y_indptr,
                                        int num_segments) {
  int seg = blockIdx.x;
  if (seg >= num_segments) return;
  int64_t xs = x_indptr[seg], xe = x_indptr[seg + 1];
  int64_t ys = y_indptr[seg];
  int64_t nbytes = (xe - xs + 7) / 8;
  for (int64_t b = threadIdx.x; b < nbytes; b += blockDim.x) {
    uint8_t v = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int64_t i = xs + b * 8 + j;
      if (i < xe && x[i]) v |= (1u << j);
    }
    y[ys + b] = v;
  }
}

}  // namespace fi

extern "C" hipError_t fi_topk(const float* x, float* out_v, int32_t* out_i, int rows,
                              int d, int k, int64_t stride, hipStream_t stream) {
  int grid = rows < 1024 ? rows : 1024;
  hipLaunchKernelGGL(fi::topk_kernel, dim3(grid), dim3(fi::TB), 0, stream, x, out_v,
                     out_i, rows, d, k, stride);
  return hipGetLastError();
}

extern "C" hipError_t fi_packbits(const uint8_t* x, uint8_t* y, int64_t n,
                                  hipStream_t stream) {
  int64_t nbytes = (n + 7) / 8;
  int grid = (int)((nbytes + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(fi::packbits_kernel, dim3(grid), dim3(256), 0, stream, x, y, n);
  return hipGetLastError();
}

extern "C" hipError_t fi_segment_packbits(const uint8_t* x, uint8_t* y,
                                          const int32_t* x_indptr,
                                          const int32_t* y_indptr, int num_segments,
                                          hipStream_t stream) {
  if (num_segments == 0) return hipSuccess;
  hipLaunchKernelGGL(fi::segment_packbits_kernel, dim3(num_segments), dim3(256), 0,
                     stream, x, y, x_indptr, y_indptr, num_segments);
  return hipGetLastError();
}
