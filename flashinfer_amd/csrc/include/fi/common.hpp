// Common device/host utilities for the MI355X (gfx950, CDNA4) kernel library.
//
// Design notes: CDNA4 wavefront is 64 lanes; LDS is 160 KiB/CU with 32 x 4 B
// banks; MFMA accumulates into the unified VGPR/AGPR file. All kernels in this
// library are written directly for gfx950 — no CUDA compatibility layer.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>
#include <stdint.h>

namespace fi {

constexpr int kWaveSize = 64;        // CDNA4 wavefront
constexpr int kNumXCD = 8;           // MI355X: 8 XCDs, private L2 each
constexpr int kNumCU = 256;          // MI355X CU count
constexpr int kLdsBytes = 160 * 1024;

using bf16 = __hip_bfloat16;
using fp16 = __half;
using fp8_e4m3 = __hip_fp8_e4m3;     // OCP e4m3fn (gfx950), NOT fnuz

// ---- CDNA4 v_permlane32_swap_b32 (guide T12): one VALU op exchanges
// register halves across lane<32 / lane>=32 — replaces a ds_bpermute
// (__shfl_xor(x, 32)) round trip through LDS. Probed semantics
// (scripts/probe/plprobe.hip): returns {concat(a.lo32, b.lo32),
// concat(a.hi32, b.hi32)}. Must run in uniform control flow. ----
__device__ __forceinline__ void permlane32_pair(uint32_t a, uint32_t b,
                                                uint32_t& lo, uint32_t& hi) {
  auto r = __builtin_amdgcn_permlane32_swap((int)a, (int)b, false, false);
  lo = (uint32_t)r[0];
  hi = (uint32_t)r[1];
}

// fmaxf(x, __shfl_xor(x, 32, 64)) without LDS
__device__ __forceinline__ float xhalf_max(float x) {
  auto r = __builtin_amdgcn_permlane32_swap(__float_as_int(x),
                                            __float_as_int(x), false, false);
  return fmaxf(__int_as_float(r[0]), __int_as_float(r[1]));
}

// x + __shfl_xor(x, 32, 64) without LDS
__device__ __forceinline__ float xhalf_sum(float x) {
  auto r = __builtin_amdgcn_permlane32_swap(__float_as_int(x),
                                            __float_as_int(x), false, false);
  return __int_as_float(r[0]) + __int_as_float(r[1]);
}
using fp8_e5m2 = __hip_fp8_e5m2;

// Ext-vector typedefs (register-resident fragments and vector loads).
typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(16))) float floatx16;
typedef __attribute__((ext_vector_type(2))) short shortx2;
typedef __attribute__((ext_vector_type(4))) short shortx4;   // 4 bf16/fp16
typedef __attribute__((ext_vector_type(8))) short shortx8;   // 8 bf16/fp16 = 16 B
typedef __attribute__((ext_vector_type(4))) int intx4;
typedef __attribute__((ext_vector_type(2))) int intx2;
typedef __attribute__((ext_vector_type(4))) unsigned int uintx4;

__device__ __forceinline__ float bf16_to_float(uint16_t u) {
  union { uint32_t i; float f; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t float_to_bf16(float f) {
  union { float f; uint32_t i; } v{f};
  // round-to-nearest-even
  uint32_t rounding_bias = 0x7FFF + ((v.i >> 16) & 1);
  return uint16_t((v.i + rounding_bias) >> 16);
}

template <typename T>
__device__ __forceinline__ float to_f32(T x);
template <>
__device__ __forceinline__ float to_f32<bf16>(bf16 x) { return __bfloat162float(x); }
template <>
__device__ __forceinline__ float to_f32<fp16>(fp16 x) { return __half2float(x); }
template <>
__device__ __forceinline__ float to_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ float to_f32<fp8_e4m3>(fp8_e4m3 x) { return (float)x; }

template <typename T>
__device__ __forceinline__ T from_f32(float x);
template <>
__device__ __forceinline__ bf16 from_f32<bf16>(float x) { return __float2bfloat16(x); }
template <>
__device__ __forceinline__ fp16 from_f32<fp16>(float x) { return __float2half(x); }
template <>
__device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ fp8_e4m3 from_f32<fp8_e4m3>(float x) { return fp8_e4m3(x); }

__host__ __device__ __forceinline__ constexpr int ceil_div(int a, int b) {
  return (a + b - 1) / b;
}
__host__ __device__ __forceinline__ constexpr int64_t ceil_div64(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

// Wave-level reductions (64-lane). width must be a power of two <= 64.
template <int width>
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = width / 2; off > 0; off >>= 1) {
    // the lane^32 exchange has a pure-VALU instruction on CDNA4; smaller
    // offsets lower to ds_swizzle and stay as-is
    if (off == 32) x = xhalf_sum(x);
    else x += __shfl_xor(x, off, 64);
  }
  return x;
}
template <int width>
__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = width / 2; off > 0; off >>= 1) {
    if (off == 32) x = xhalf_max(x);
    else x = fmaxf(x, __shfl_xor(x, off, 64));
  }
  return x;
}

// XCD-aware workgroup remap (bijective, any grid size): consecutive output
// ids land on one XCD so neighboring tiles share its private L2.
// See MI355X guide §5 T1/ERRATA#11.
__device__ __forceinline__ uint32_t xcd_swizzle(uint32_t bid, uint32_t nwg) {
  if (nwg < kNumXCD * 2) return bid;
  uint32_t xcd = bid % kNumXCD, idx = bid / kNumXCD;
  uint32_t q = nwg / kNumXCD, r = nwg % kNumXCD;
  uint32_t base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + idx;
}

#define FI_HIP_CHECK(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      return _e;                                                             \
    }                                                                        \
  } while (0)

}  // namespace fi
