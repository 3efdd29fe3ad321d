// High-throughput bf16 NT GEMM for gfx950 — 256x256 tile, BK=64, 8 waves,
// double-buffered LDS staged with __builtin_amdgcn_global_load_lds (16 B
// direct-to-LDS, the single biggest staging lever on CDNA4), XOR-swizzled
// via pre-swizzled global source addresses (global_load_lds writes linearly;
// the source permutation and the ds_read permutation are the same
// involution), counted-vmcnt software pipeline with s_setprio around the
// MFMA cluster (guide §5 template / T1-T5).
//
// C[M,N] = A[M,K] x B[N,K]^T, bf16 in / f32 accumulate / bf16 out.
#include "fi/common.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"

namespace fi {

namespace gemm2 {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTH = 512;  // 8 waves: 2 (M) x 4 (N)
constexpr int WM = 128, WN = 64;  // per-wave output

// LDS tiles: [256][64] bf16 = 32 KB each, double buffered -> 128 KB total.
// Rows are 128 B -> swz128 spreads the 8 16-B slots.

__device__ __forceinline__ void stage_tile(const bf16* __restrict__ gbase, int64_t ld,
                                           int row0, int rows_max, int k0, int K,
                                           uint32_t lds_base_bytes, int tid) {
  // 2048 16-B units; thread t issues units t, t+512, t+1024, t+1536 so each
  // wave's 64 units are contiguous (global_load_lds dest = uniform base +
  // lane*16).
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int u = tid + i * NTH;
    uint32_t dst_byte = (uint32_t)u * 16;
    uint32_t logical = swz128(dst_byte);  // involution: source for this dest
    int row = logical >> 7;
    int col = (logical & 127) >> 1;
    int gm = row0 + row;
    int gk = k0 + col;
    // clamp OOB to row0/k0 (loads garbage inside the tensor; masked later by
    // the K-loop bounds and epilogue bounds — full tiles take the fast path)
    const bf16* src = gbase + (int64_t)(gm < rows_max ? gm : rows_max - 1) * ld +
                      (gk < K ? gk : 0);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                     (__attribute__((address_space(3))) void*)(uintptr_t)(
                                         lds_base_bytes + dst_byte),
                                     16, 0, 0);
  }
}

__global__ __launch_bounds__(NTH, 1) void gemm_nt_v2_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B, bf16* __restrict__ C,
    int M, int N, int K, int64_t lda, int64_t ldb, int64_t ldc, float alpha) {
  __shared__ bf16 As[2][BM * BK];
  __shared__ bf16 Bs[2][BN * BK];

  uint32_t nwg = gridDim.x;
  uint32_t wg = xcd_swizzle(blockIdx.x, nwg);
  int tiles_n = (N + BN - 1) / BN;
  int bm0 = (int)(wg / tiles_n) * BM;
  int bn0 = (int)(wg % tiles_n) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 2) * WM;  // 0 or 128
  const int wn = (wid & 3) * WN;   // 0,64,128,192
  const int line = lane & 31;
  const int khalf = (lane >> 5) * 8;

  floatx16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {};

  const uint32_t as_base = (uint32_t)(uintptr_t)&As[0][0];
  const uint32_t bs_base = (uint32_t)(uintptr_t)&Bs[0][0];
  constexpr uint32_t BUF_BYTES = BM * BK * 2;

  int nk = (K + BK - 1) / BK;
  // prologue: stage k-tile 0 into buffer 0
  stage_tile(A, lda, bm0, M, 0, K, as_base, tid);
  stage_tile(B, ldb, bn0, N, 0, K, bs_base, tid);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < nk; ++kt) {
    // issue next-tile staging FIRST (overlaps with this tile's compute)
    if (kt + 1 < nk) {
      stage_tile(A, lda, bm0, M, (kt + 1) * BK, K, as_base + (cur ^ 1) * BUF_BYTES,
                 tid);
      stage_tile(B, ldb, bn0, N, (kt + 1) * BK, K, bs_base + (cur ^ 1) * BUF_BYTES,
                 tid);
    }
    const char* a_lds = (const char*)&As[cur][0];
    const char* b_lds = (const char*)&Bs[cur][0];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BK / 16; ++ks) {
      bf16x8_t af[4], bf[2];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = *reinterpret_cast<const bf16x8_t*>(
            a_lds + swz128((wm + i * 32 + line) * (BK * 2) + (ks * 16 + khalf) * 2));
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bf[j] = *reinterpret_cast<const bf16x8_t*>(
            b_lds + swz128((wn + j * 32 + line) * (BK * 2) + (ks * 16 + khalf) * 2));
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = mfma_32x32x16_bf16(af[i], bf[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int m = bm0 + wm + i * 32 + mfma32_cd_row(r, lane);
        int n = bn0 + wn + j * 32 + mfma32_cd_col(lane);
        if (m < M && n < N) C[(int64_t)m * ldc + n] = from_f32<bf16>(acc[i][j][r] * alpha);
      }
    }
  }
}

}  // namespace gemm2

}  // namespace fi

extern "C" hipError_t fi_gemm_nt_v2(const void* A, const void* B, void* C, int M, int N,
                                    int K, int64_t lda, int64_t ldb, int64_t ldc,
                                    float alpha, hipStream_t stream) {
  int tiles = ((M + fi::gemm2::BM - 1) / fi::gemm2::BM) *
              ((N + fi::gemm2::BN - 1) / fi::gemm2::BN);
  hipLaunchKernelGGL(fi::gemm2::gemm_nt_v2_kernel, dim3(tiles), dim3(fi::gemm2::NTH), 0,
                     stream, (const fi::bf16*)A, (const fi::bf16*)B, (fi::bf16*)C, M, N,
                     K, lda, ldb, ldc, alpha);
  return hipGetLastError();
}
