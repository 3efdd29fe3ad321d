// RMSNorm / LayerNorm family for gfx950. Functional parity with the
// reference's include/flashinfer/norm.cuh (RMSNormKernel:63,
// FusedAddRMSNormKernel:414, Gemma variants:679, generalLayerNorm:763) but
// written CDNA4-first: 256-thread (4-wave) workgroups, bf16x8 16-byte lane
// loads (hipcc does not auto-vectorize bf16), f32 accumulation, LDS+shuffle
// block reduction, grid-stride over rows capped for the 256-CU chip.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int kNormThreads = 256;
constexpr int kNormWaves = kNormThreads / kWaveSize;

template <int NW>
__device__ __forceinline__ float block_reduce_sum(float x, float* smem) {
  x = wave_reduce_sum<kWaveSize>(x);
  int wave = threadIdx.x / kWaveSize;
  int lane = threadIdx.x % kWaveSize;
  if (lane == 0) smem[wave] = x;
  __syncthreads();
  float r = (lane < NW) ? smem[lane] : 0.f;
  r = wave_reduce_sum<NW>(r);
  // broadcast via smem slot NW
  if (threadIdx.x == 0) smem[NW] = r;
  __syncthreads();
  r = smem[NW];
  __syncthreads();
  return r;
}

// ---------------- RMSNorm ----------------
// y = x / rms(x) * (w + wbias)    (optionally y = silu(...) — reference
// flashinfer/norm/__init__.py fused_rmsnorm_silu:689)
template <typename T, int VEC, bool kWeightBias, bool kSilu = false>
__global__ void rmsnorm_kernel(const T* __restrict__ x, const T* __restrict__ w,
                               T* __restrict__ y, int rows, int d, int64_t stride_x,
                               int64_t stride_y, float eps) {
  __shared__ float smem[kNormWaves + 1];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (int64_t)row * stride_x;
    T* yr = y + (int64_t)row * stride_y;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v;
      v.load(xr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = v.get(j);
        ss += f * f;
      }
    }
    ss = block_reduce_sum<kNormWaves>(ss, smem);
    float rrms = rsqrtf(ss / d + eps);
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v, wv, out;
      v.load(xr + i);
      wv.load(w + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float wj = kWeightBias ? wv.get(j) + 1.f : wv.get(j);
        float r = v.get(j) * rrms * wj;
        if constexpr (kSilu) r *= 1.f / (1.f + __builtin_expf(-r));
        out.set(j, r);
      }
      out.store(yr + i);
    }
    __syncthreads();
  }
}

// ---------------- RMSNorm + fp8 quantize ----------------
// Reference flashinfer/norm/__init__.py rmsnorm_quant:214 /
// fused_add_rmsnorm_quant:323:  out = (rmsnorm(x or x+residual) * w) / scale
// cast to fp8 e4m3; with kAdd the residual is updated in place first.
template <typename T, int VEC, bool kAdd>
__global__ void rmsnorm_quant_kernel(T* __restrict__ x, T* __restrict__ residual,
                                     const T* __restrict__ w,
                                     fp8_e4m3* __restrict__ out,
                                     const float* __restrict__ scale, int rows, int d,
                                     float eps) {
  __shared__ float smem[kNormWaves + 1];
  const float inv_scale = 1.f / *scale;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    T* xr = x + (int64_t)row * d;
    T* rr = kAdd ? residual + (int64_t)row * d : nullptr;
    fp8_e4m3* orow = out + (int64_t)row * d;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> vx;
      vx.load(xr + i);
      if constexpr (kAdd) {
        vec_t<T, VEC> vr;
        vr.load(rr + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = vx.get(j) + vr.get(j);
          vr.set(j, f);
          ss += f * f;
        }
        vr.store(rr + i);
      } else {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = vx.get(j);
          ss += f * f;
        }
      }
    }
    ss = block_reduce_sum<kNormWaves>(ss, smem);
    float rrms = rsqrtf(ss / d + eps);
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v, wv;
      v.load(kAdd ? rr + i : xr + i);
      wv.load(w + i);
      vec_t<fp8_e4m3, VEC> q;
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        q.set(j, v.get(j) * rrms * wv.get(j) * inv_scale);
      q.store(orow + i);
    }
    __syncthreads();
  }
}

// ---------------- Fused Add + RMSNorm ----------------
// residual += x;  x = rmsnorm(residual) * (w + wbias)   (both in-place)
template <typename T, int VEC, bool kWeightBias>
__global__ void fused_add_rmsnorm_kernel(T* __restrict__ x, T* __restrict__ residual,
                                         const T* __restrict__ w, int rows, int d,
                                         int64_t stride_x, int64_t stride_r, float eps) {
  __shared__ float smem[kNormWaves + 1];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    T* xr = x + (int64_t)row * stride_x;
    T* rr = residual + (int64_t)row * stride_r;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> vx, vr;
      vx.load(xr + i);
      vr.load(rr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = vx.get(j) + vr.get(j);
        vr.set(j, f);
        ss += f * f;
      }
      vr.store(rr + i);
    }
    ss = block_reduce_sum<kNormWaves>(ss, smem);
    float rrms = rsqrtf(ss / d + eps);
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> vr, wv, out;
      vr.load(rr + i);
      wv.load(w + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float wj = kWeightBias ? wv.get(j) + 1.f : wv.get(j);
        out.set(j, vr.get(j) * rrms * wj);
      }
      out.store(xr + i);
    }
    __syncthreads();
  }
}

// ---------------- LayerNorm + fp8 quantize ----------------
// out = ((x - mean) / sqrt(var + eps) * w (+ b)) / scale -> fp8 e4m3
// (reference flashinfer/norm/__init__.py layernorm_quant role)
template <typename T, int VEC>
__global__ void layernorm_quant_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                       const T* __restrict__ b,
                                       fp8_e4m3* __restrict__ out,
                                       const float* __restrict__ scale, int rows,
                                       int d, float eps) {
  __shared__ float smem[kNormWaves + 1];
  const float inv_scale = 1.f / *scale;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (int64_t)row * d;
    fp8_e4m3* orow = out + (int64_t)row * d;
    float sum = 0.f, sq = 0.f;
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v;
      v.load(xr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = v.get(j);
        sum += f;
        sq += f * f;
      }
    }
    sum = block_reduce_sum<kNormWaves>(sum, smem);
    sq = block_reduce_sum<kNormWaves>(sq, smem);
    float mean = sum / d;
    float rstd = rsqrtf(sq / d - mean * mean + eps);
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v, wv, bv;
      v.load(xr + i);
      wv.load(w + i);
      if (b) bv.load(b + i);
      vec_t<fp8_e4m3, VEC> q;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float r = (v.get(j) - mean) * rstd * wv.get(j);
        if (b) r += bv.get(j);
        q.set(j, r * inv_scale);
      }
      q.store(orow + i);
    }
    __syncthreads();
  }
}

// ---------------- LayerNorm ----------------
template <typename T, int VEC>
__global__ void layernorm_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                 const T* __restrict__ b, T* __restrict__ y, int rows,
                                 int d, int64_t stride_x, int64_t stride_y, float eps) {
  __shared__ float smem[kNormWaves + 1];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (int64_t)row * stride_x;
    T* yr = y + (int64_t)row * stride_y;
    float sum = 0.f, sq = 0.f;
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v;
      v.load(xr + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = v.get(j);
        sum += f;
        sq += f * f;
      }
    }
    sum = block_reduce_sum<kNormWaves>(sum, smem);
    sq = block_reduce_sum<kNormWaves>(sq, smem);
    float mean = sum / d;
    float var = sq / d - mean * mean;
    float rstd = rsqrtf(var + eps);
    for (int i = threadIdx.x * VEC; i < d; i += kNormThreads * VEC) {
      vec_t<T, VEC> v, wv, bv, out;
      v.load(xr + i);
      wv.load(w + i);
      if (b) bv.load(b + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float r = (v.get(j) - mean) * rstd * wv.get(j);
        if (b) r += bv.get(j);
        out.set(j, r);
      }
      out.store(yr + i);
    }
    __syncthreads();
  }
}

template <typename T>
hipError_t norm_launch(int which, const void* x, const void* w, const void* b, void* y,
                       void* residual, int rows, int d, int64_t sx, int64_t sy, float eps,
                       bool weight_bias, hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  dim3 g(grid), blk(kNormThreads);
  bool vec8 = (d % 8 == 0);
  auto X = (const T*)x;
  auto W = (const T*)w;
  auto B = (const T*)b;
  auto Y = (T*)y;
  auto R = (T*)residual;
#define LAUNCH_N(kern, VEC, WB, ...) \
  hipLaunchKernelGGL((kern<T, VEC, WB>), g, blk, 0, stream, __VA_ARGS__)
  switch (which) {
    case 0:  // rmsnorm
      if (vec8) {
        if (weight_bias) LAUNCH_N(rmsnorm_kernel, 8, true, X, W, Y, rows, d, sx, sy, eps);
        else LAUNCH_N(rmsnorm_kernel, 8, false, X, W, Y, rows, d, sx, sy, eps);
      } else {
        if (weight_bias) LAUNCH_N(rmsnorm_kernel, 1, true, X, W, Y, rows, d, sx, sy, eps);
        else LAUNCH_N(rmsnorm_kernel, 1, false, X, W, Y, rows, d, sx, sy, eps);
      }
      break;
    case 1:  // fused add rmsnorm (x==input/out, y unused, residual in/out)
      if (vec8) {
        if (weight_bias) LAUNCH_N(fused_add_rmsnorm_kernel, 8, true, (T*)x, R, W, rows, d, sx, sy, eps);
        else LAUNCH_N(fused_add_rmsnorm_kernel, 8, false, (T*)x, R, W, rows, d, sx, sy, eps);
      } else {
        if (weight_bias) LAUNCH_N(fused_add_rmsnorm_kernel, 1, true, (T*)x, R, W, rows, d, sx, sy, eps);
        else LAUNCH_N(fused_add_rmsnorm_kernel, 1, false, (T*)x, R, W, rows, d, sx, sy, eps);
      }
      break;
    case 2:  // layernorm
      if (vec8) hipLaunchKernelGGL((layernorm_kernel<T, 8>), g, blk, 0, stream, X, W, B, Y, rows, d, sx, sy, eps);
      else hipLaunchKernelGGL((layernorm_kernel<T, 1>), g, blk, 0, stream, X, W, B, Y, rows, d, sx, sy, eps);
      break;
    case 3:  // fused rmsnorm + silu
      if (vec8) hipLaunchKernelGGL((rmsnorm_kernel<T, 8, false, true>), g, blk, 0, stream, X, W, Y, rows, d, sx, sy, eps);
      else hipLaunchKernelGGL((rmsnorm_kernel<T, 1, false, true>), g, blk, 0, stream, X, W, Y, rows, d, sx, sy, eps);
      break;
  }
#undef LAUNCH_N
  return hipGetLastError();
}

template <typename T>
hipError_t norm_quant_launch(bool add, void* x, void* residual, const void* w, void* out,
                             const float* scale, int rows, int d, float eps,
                             hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  dim3 g(grid), blk(kNormThreads);
  bool vec8 = (d % 8 == 0);
  auto X = (T*)x;
  auto R = (T*)residual;
  auto W = (const T*)w;
  auto O = (fp8_e4m3*)out;
  if (add) {
    if (vec8) hipLaunchKernelGGL((rmsnorm_quant_kernel<T, 8, true>), g, blk, 0, stream, X, R, W, O, scale, rows, d, eps);
    else hipLaunchKernelGGL((rmsnorm_quant_kernel<T, 1, true>), g, blk, 0, stream, X, R, W, O, scale, rows, d, eps);
  } else {
    if (vec8) hipLaunchKernelGGL((rmsnorm_quant_kernel<T, 8, false>), g, blk, 0, stream, X, R, W, O, scale, rows, d, eps);
    else hipLaunchKernelGGL((rmsnorm_quant_kernel<T, 1, false>), g, blk, 0, stream, X, R, W, O, scale, rows, d, eps);
  }
  return hipGetLastError();
}

}  // namespace fi

namespace fi {
template <typename T>
hipError_t ln_quant_launch(const void* x, const void* w, const void* b, void* out,
                           const float* scale, int rows, int d, float eps,
                           hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  dim3 g(grid), blk(kNormThreads);
  if (d % 8 == 0)
    hipLaunchKernelGGL((layernorm_quant_kernel<T, 8>), g, blk, 0, stream, (const T*)x,
                       (const T*)w, (const T*)b, (fp8_e4m3*)out, scale, rows, d, eps);
  else
    hipLaunchKernelGGL((layernorm_quant_kernel<T, 1>), g, blk, 0, stream, (const T*)x,
                       (const T*)w, (const T*)b, (fp8_e4m3*)out, scale, rows, d, eps);
  return hipGetLastError();
}
}  // namespace fi

extern "C" hipError_t fi_layernorm_quant(int dtype, const void* x, const void* w,
                                         const void* b, void* out, const float* scale,
                                         int rows, int d, float eps,
                                         hipStream_t stream) {
  switch (dtype) {
    case 0: return fi::ln_quant_launch<fi::bf16>(x, w, b, out, scale, rows, d, eps, stream);
    case 1: return fi::ln_quant_launch<fi::fp16>(x, w, b, out, scale, rows, d, eps, stream);
    case 2: return fi::ln_quant_launch<float>(x, w, b, out, scale, rows, d, eps, stream);
  }
  return hipErrorInvalidValue;
}

extern "C" hipError_t fi_norm_quant(int add, int dtype, void* x, void* residual,
                                    const void* w, void* out, const float* scale,
                                    int rows, int d, float eps, hipStream_t stream) {
  switch (dtype) {
    case 0: return fi::norm_quant_launch<fi::bf16>(add, x, residual, w, out, scale, rows, d, eps, stream);
    case 1: return fi::norm_quant_launch<fi::fp16>(add, x, residual, w, out, scale, rows, d, eps, stream);
    case 2: return fi::norm_quant_launch<float>(add, x, residual, w, out, scale, rows, d, eps, stream);
  }
  return hipErrorInvalidValue;
}

// which: 0 rmsnorm, 1 fused_add_rmsnorm, 2 layernorm, 3 rmsnorm+silu
// dtype: 0 bf16, 1 fp16, 2 fp32
extern "C" hipError_t fi_norm(int which, int dtype, const void* x, const void* w,
                              const void* b, void* y, void* residual, int rows, int d,
                              int64_t sx, int64_t sy, float eps, int weight_bias,
                              hipStream_t stream) {
  switch (dtype) {
    case 0: return fi::norm_launch<fi::bf16>(which, x, w, b, y, residual, rows, d, sx, sy, eps, weight_bias, stream);
    case 1: return fi::norm_launch<fi::fp16>(which, x, w, b, y, residual, rows, d, sx, sy, eps, weight_bias, stream);
    case 2: return fi::norm_launch<float>(which, x, w, b, y, residual, rows, d, sx, sy, eps, weight_bias, stream);
  }
  return hipErrorInvalidValue;
}
