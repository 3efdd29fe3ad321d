// MFMA wrappers + fragment layout helpers for gfx950 (CDNA4).
//
// Shapes used in this library (all per-wave, accumulate f32):
//   v_mfma_f32_16x16x32_bf16 : A 16x32, B 32x16, C/D 16x16 (4 f32/lane)
//   v_mfma_f32_32x32x16_bf16 : A 32x16, B 16x32, C/D 32x32 (16 f32/lane)
//   v_mfma_f32_16x16x32_fp8_fp8 / 32x32x16_fp8_fp8 (OCP e4m3)
//
// Operand layouts (AMD matrix-core mapping, verified in tests/gpu):
//   16x16x32 A: row = lane & 15, k = (lane >> 4) * 8 + j     (j in [0,8))
//   16x16x32 B: col = lane & 15, k = (lane >> 4) * 8 + j
//   16x16   C/D: col = lane & 15, row = (lane >> 4) * 4 + r  (r in [0,4))
//   32x32x16 A: row = lane & 31, k = (lane >> 5) * 8 + j
//   32x32x16 B: col = lane & 31, k = (lane >> 5) * 8 + j
//   32x32   C/D: col = lane & 31, row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5)
//               (r in [0,16))
#pragma once
#include "fi/common.hpp"

namespace fi {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8_t;

// ---- bf16 ----
__device__ __forceinline__ floatx4 mfma_16x16x32_bf16(bf16x8_t a, bf16x8_t b, floatx4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ floatx16 mfma_32x32x16_bf16(bf16x8_t a, bf16x8_t b, floatx16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// ---- fp16 ----
__device__ __forceinline__ floatx4 mfma_16x16x32_f16(f16x8_t a, f16x8_t b, floatx4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ floatx16 mfma_32x32x16_f16(f16x8_t a, f16x8_t b, floatx16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
}

// ---- fp8 (OCP e4m3fn). Operands are 8 packed fp8 bytes per lane = i64. ----
__device__ __forceinline__ floatx4 mfma_16x16x32_fp8(int64_t a, int64_t b, floatx4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ floatx16 mfma_32x32x16_fp8(int64_t a, int64_t b, floatx16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(a, b, c, 0, 0, 0);
}

// gfx950 large-K fp8 via the block-scaled f8f6f4 instruction with unit e8m0
// scales (0x7F = 2^0): 2x the non-scaled fp8 rate (guide: 4686 vs 2190 TF).
// Fragments: 32 contiguous K bytes per lane — A row = lane&31,
// k = (lane>>5)*32 + j; B col = lane&31, same k; C/D as every 32x32 shape.
// [probed on MI355X: scripts/probe/mfma64.hip, maxerr 0 vs fp32 reference]
typedef __attribute__((ext_vector_type(8))) int intx8;
__device__ __forceinline__ floatx16 mfma_32x32x64_fp8(intx8 a, intx8 b, floatx16 c) {
  return __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, 127, 0,
                                                         127);
}

// Generic dtype dispatch for attention/GEMM templates.
template <typename T>
struct mfma_ab_frag;
template <>
struct mfma_ab_frag<bf16> {
  using type = bf16x8_t;
  __device__ static floatx4 mma16(type a, type b, floatx4 c) { return mfma_16x16x32_bf16(a, b, c); }
  __device__ static floatx16 mma32(type a, type b, floatx16 c) { return mfma_32x32x16_bf16(a, b, c); }
};
template <>
struct mfma_ab_frag<fp16> {
  using type = f16x8_t;
  __device__ static floatx4 mma16(type a, type b, floatx4 c) { return mfma_16x16x32_f16(a, b, c); }
  __device__ static floatx16 mma32(type a, type b, floatx16 c) { return mfma_32x32x16_f16(a, b, c); }
};

}  // namespace fi
