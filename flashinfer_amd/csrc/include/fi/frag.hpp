// MFMA fragment addressing helpers for gfx950 (shared by GEMM and attention).
//
// For v_mfma_f32_32x32x16_{bf16,f16} (D = A[32x16] * B[16x32] + C[32x32]):
//   A: lane holds A[row = lane&31][k = (lane>>5)*8 + j], j in [0,8)
//   B: lane holds B[k = (lane>>5)*8 + j][col = lane&31]
//   C/D: lane holds C[row = (r&3) + 8*(r>>2) + 4*(lane>>5)][col = lane&31],
//        r in [0,16)
// For v_mfma_f32_16x16x32_{bf16,f16} (D = A[16x32] * B[32x16] + C[16x16]):
//   A: lane holds A[row = lane&15][k = (lane>>4)*8 + j]
//   B: lane holds B[k = (lane>>4)*8 + j][col = lane&15]
//   C/D: lane holds C[row = (lane>>4)*4 + r][col = lane&15], r in [0,4)
// (layouts verified on-device by tests/test_gemm.py asymmetric refchecks)
#pragma once
#include "fi/common.hpp"
#include "fi/mfma.hpp"

namespace fi {

// --- 32x32x16 C/D element coordinates ---
__device__ __forceinline__ int mfma32_cd_row(int r, int lane) {
  return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
}
__device__ __forceinline__ int mfma32_cd_col(int lane) { return lane & 31; }

// --- 16x16x32 C/D element coordinates ---
__device__ __forceinline__ int mfma16_cd_row(int r, int lane) {
  return (lane >> 4) * 4 + r;
}
__device__ __forceinline__ int mfma16_cd_col(int lane) { return lane & 15; }

// A/B fragment source coordinates for 32x32x16 (row/col index in tile, k0 =
// base of the 16-wide k slice): each lane reads 8 contiguous k elements
// starting at k0 + (lane>>5)*8 from row (lane&31) — a single 16-byte load
// when the tile is k-contiguous.
__device__ __forceinline__ int mfma32_ab_line(int lane) { return lane & 31; }
__device__ __forceinline__ int mfma32_ab_k(int lane) { return (lane >> 5) * 8; }

__device__ __forceinline__ int mfma16_ab_line(int lane) { return lane & 15; }
__device__ __forceinline__ int mfma16_ab_k(int lane) { return (lane >> 4) * 8; }

// XOR swizzle for row-major LDS tiles with 128-byte rows (e.g. [*][64] bf16):
// spreads the 8-slot 16B structure across banks; apply to BOTH write and read
// byte offsets (guide §6 G4).
__device__ __forceinline__ uint32_t swz128(uint32_t byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}
// For 256-byte rows ([*][128] bf16): 16 16-byte slots per row — XOR with 4
// row bits for the full-slot spread (8-slot spread leaves rows r and r+8
// aliased -> 4-way conflicts on 32-row fragment reads).
__device__ __forceinline__ uint32_t swz256(uint32_t byte_off) {
  return byte_off ^ (((byte_off >> 8) & 15) << 4);
}

}  // namespace fi
