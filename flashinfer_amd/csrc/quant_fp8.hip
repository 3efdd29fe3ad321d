// FP8 (OCP e4m3fn — gfx950 native, NOT the MI300X fnuz variant) quantization
// kernels. Parity with reference fp8_quantization.py / nv_internal thop quant
// ops: per-token-group (1x128) activation quant and per-tensor quant.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr float kF8Max = 448.f;  // e4m3fn max

__device__ __forceinline__ uint8_t to_fp8(float x) {
  return __hip_cvt_float_to_fp8(x, __HIP_SATFINITE, __HIP_E4M3);
}

// Per-128-group quantize: a wave covers FOUR groups (512 elems; lane holds
// 8 contiguous values via a 16B load, group amax is a segmented 16-lane
// shfl tree, 8 fp8 bytes store as one u64). The 2-elem-per-lane version
// measured 2.5 TB/s in the MoE loop (profiles/README r02 addendum 10).
// Rows with K % 512 fall back to one-wave-per-group for the tail groups.
// scale layout: MN-major [K/128, M] (transposed=true) or [M, K/128].
template <typename T, bool TRANS_SCALE>
__global__ void per_group_quant_kernel(const T* __restrict__ x, uint8_t* __restrict__ q,
                                       float* __restrict__ scale, int64_t rows, int K,
                                       int64_t stride_row, float eps) {
  const int groups = K / 128;
  const int c4 = K / 512;                 // full 4-group wave clusters
  const int tail = groups - c4 * 4;       // leftover 128-groups
  const int64_t total = rows * (c4 + tail);
  int64_t wave = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  if (wave >= total) return;
  int item = (int)(wave % (c4 + tail));
  int64_t row = wave / (c4 + tail);
  if (item < c4) {
    const T* src = x + row * stride_row + item * 512;
    vec_t<T, 8> lv;
    lv.load(src + lane * 8);
    float v[8];
    float amax = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      v[j] = lv.get(j);
      amax = fmaxf(amax, fabsf(v[j]));
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, 16));
    float s = fmaxf(amax, eps) / kF8Max;
    float inv = 1.f / s;
    uint64_t packed = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      packed |= (uint64_t)__builtin_bit_cast(uint8_t, to_fp8(v[j] * inv))
                << (8 * j);
    *reinterpret_cast<uint64_t*>(q + row * K + item * 512 + lane * 8) = packed;
    if ((lane & 15) == 0) {
      int g = item * 4 + (lane >> 4);
      if constexpr (TRANS_SCALE) scale[(int64_t)g * rows + row] = s;
      else scale[row * groups + g] = s;
    }
  } else {
    int g = c4 * 4 + (item - c4);
    const T* src = x + row * stride_row + g * 128;
    float v[2];
    vec_t<T, 2> lv;
    lv.load(src + lane * 2);
    v[0] = lv.get(0);
    v[1] = lv.get(1);
    float amax = fmaxf(fabsf(v[0]), fabsf(v[1]));
    amax = wave_reduce_max<kWaveSize>(amax);
    float s = fmaxf(amax, eps) / kF8Max;
    float inv = 1.f / s;
    uint8_t* dst = q + row * K + g * 128 + lane * 2;
    dst[0] = to_fp8(v[0] * inv);
    dst[1] = to_fp8(v[1] * inv);
    if (lane == 0) {
      if constexpr (TRANS_SCALE) scale[(int64_t)g * rows + row] = s;
      else scale[row * groups + g] = s;
    }
  }
}

template <typename T>
__global__ void scale_quant_kernel(const T* __restrict__ x, uint8_t* __restrict__ q,
                                   const float* __restrict__ inv_scale, int64_t n) {
  float is = *inv_scale;
  // vectorized main body: 16B loads / 8B stores per thread
  int64_t n8 = n / 8;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    vec_t<T, 8> lv;
    lv.load(x + i * 8);
    uint64_t packed = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      packed |= (uint64_t)__builtin_bit_cast(uint8_t, to_fp8(lv.get(j) * is))
                << (8 * j);
    *reinterpret_cast<uint64_t*>(q + i * 8) = packed;
  }
  for (int64_t i = n8 * 8 + blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n; i += (int64_t)gridDim.x * blockDim.x) {
    q[i] = to_fp8(to_f32<T>(x[i]) * is);
  }
}

}  // namespace fi

extern "C" hipError_t fi_per_group_quant_fp8(int dtype, int trans_scale, const void* x,
                                             void* q, float* scale, int64_t rows, int K,
                                             int64_t stride_row, float eps,
                                             hipStream_t stream) {
  if (K % 128 != 0) return hipErrorInvalidValue;
  // work item = one 4-group cluster (512 elems) or one tail 128-group
  int64_t waves = rows * ((K / 512) + (K / 128 - (K / 512) * 4));
  int64_t blocks = (waves * 64 + 255) / 256;
  dim3 g((uint32_t)(blocks < 0x7FFFFFFF ? blocks : 0x7FFFFFFF)), blk(256);
#define LQ(T, TS)                                                                  \
  hipLaunchKernelGGL((fi::per_group_quant_kernel<T, TS>), g, blk, 0, stream,        \
                     (const T*)x, (uint8_t*)q, scale, rows, K, stride_row, eps)
  switch (dtype * 2 + trans_scale) {
    case 0: LQ(fi::bf16, false); break;
    case 1: LQ(fi::bf16, true); break;
    case 2: LQ(fi::fp16, false); break;
    case 3: LQ(fi::fp16, true); break;
    case 4: LQ(float, false); break;
    case 5: LQ(float, true); break;
    default: return hipErrorInvalidValue;
  }
#undef LQ
  return hipGetLastError();
}

extern "C" hipError_t fi_scale_quant_fp8(int dtype, const void* x, void* q,
                                         const float* inv_scale, int64_t n,
                                         hipStream_t stream) {
  int grid = (int)((n + 255) / 256);
  if (grid > 2048) grid = 2048;
  dim3 g(grid), blk(256);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((fi::scale_quant_kernel<fi::bf16>), g, blk, 0, stream,
                         (const fi::bf16*)x, (uint8_t*)q, inv_scale, n);
      break;
    case 1:
      hipLaunchKernelGGL((fi::scale_quant_kernel<fi::fp16>), g, blk, 0, stream,
                         (const fi::fp16*)x, (uint8_t*)q, inv_scale, n);
      break;
    case 2:
      hipLaunchKernelGGL((fi::scale_quant_kernel<float>), g, blk, 0, stream,
                         (const float*)x, (uint8_t*)q, inv_scale, n);
      break;
    default:
      return hipErrorInvalidValue;
  }
  return hipGetLastError();
}
