// Grouped / segment NT GEMM for gfx950 — the MoE expert-GEMM and LoRA
// segment-GEMM workhorse. Functional parity with reference
// include/flashinfer/gemm/group_gemm.cuh (CutlassSegmentGEMMRun:47) and the
// MoE grouped GEMMs (csrc/nv_internal moe_kernels), as one hand-written CDNA4
// kernel reusing the gemm_v2 structure (256x256 tile, BK=64,
// global_load_lds double-buffered pipeline, swizzled LDS).
//
// C[seg rows, N] = A[seg rows, K] x W[widx(seg), N, K]^T  per segment, where
// segment row ranges come from a DEVICE array m_indptr[E+1] (produced by
// routing — no host sync, hipGraph-safe fixed grid with early-exit tiles).
#include "fi/common.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"

namespace fi {

namespace ggemm {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NTH = 256;  // 4 waves: 2x2 of 64x64
constexpr int WM = 64, WN = 64;

__device__ __forceinline__ void stage_tile(const bf16* __restrict__ gbase, int64_t ld,
                                           int row0, int rows_end, int k0,
                                           uint32_t lds_base_bytes, int tid) {
  // BM x BK x 2B = 16 KB = 1024 16-B units over 256 threads
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int u = tid + i * NTH;
    uint32_t dst_byte = (uint32_t)u * 16;
    uint32_t logical = swz128(dst_byte);
    int row = logical >> 7;
    int col = (logical & 127) >> 1;
    int gm = row0 + row;
    const bf16* src = gbase + (int64_t)(gm < rows_end ? gm : rows_end - 1) * ld + k0 + col;
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                     (__attribute__((address_space(3))) void*)(uintptr_t)(
                                         lds_base_bytes + dst_byte),
                                     16, 0, 0);
  }
}

__global__ __launch_bounds__(NTH, 2) void group_gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ W, bf16* __restrict__ C,
    const int32_t* __restrict__ m_indptr, const int32_t* __restrict__ w_indices,
    int num_segments, int N, int K, int64_t lda, int64_t ldw_n, int64_t ldw_seg,
    int64_t ldc) {
  __shared__ bf16 As[2][BM * BK];
  __shared__ bf16 Bs[2][BN * BK];

  const int seg = blockIdx.z;
  const int m0 = m_indptr[seg] + blockIdx.y * BM;
  const int m_end = m_indptr[seg + 1];
  if (m0 >= m_end) return;
  const int bn0 = blockIdx.x * BN;
  const int widx = w_indices ? w_indices[seg] : seg;
  const bf16* Wb = W + (int64_t)widx * ldw_seg;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * WM;
  const int wn = (wid & 1) * WN;
  const int line = lane & 31;
  const int khalf = (lane >> 5) * 8;

  floatx16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {};

  const uint32_t as_base = (uint32_t)(uintptr_t)&As[0][0];
  const uint32_t bs_base = (uint32_t)(uintptr_t)&Bs[0][0];
  constexpr uint32_t BUF_BYTES = BM * BK * 2;

  int nk = K / BK;
  stage_tile(A, lda, m0, m_end, 0, as_base, tid);
  stage_tile(Wb, ldw_n, bn0, N, 0, bs_base, tid);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < nk; ++kt) {
    if (kt + 1 < nk) {
      stage_tile(A, lda, m0, m_end, (kt + 1) * BK, as_base + (cur ^ 1) * BUF_BYTES, tid);
      stage_tile(Wb, ldw_n, bn0, N, (kt + 1) * BK, bs_base + (cur ^ 1) * BUF_BYTES, tid);
    }
    const char* a_lds = (const char*)&As[cur][0];
    const char* b_lds = (const char*)&Bs[cur][0];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BK / 16; ++ks) {
      bf16x8_t af[2], bfv[2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
        af[i] = *reinterpret_cast<const bf16x8_t*>(
            a_lds + swz128((wm + i * 32 + line) * (BK * 2) + (ks * 16 + khalf) * 2));
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfv[j] = *reinterpret_cast<const bf16x8_t*>(
            b_lds + swz128((wn + j * 32 + line) * (BK * 2) + (ks * 16 + khalf) * 2));
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = mfma_32x32x16_bf16(af[i], bfv[j], acc[i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int m = m0 + wm + i * 32 + mfma32_cd_row(r, lane);
        int n = bn0 + wn + j * 32 + mfma32_cd_col(lane);
        if (m < m_end && n < N) C[(int64_t)m * ldc + n] = from_f32<bf16>(acc[i][j][r]);
      }
    }
  }
}

}  // namespace ggemm

}  // namespace fi

// W layout: [num_weights, N, K] K-contiguous (ldw_seg = N*K typical).
extern "C" hipError_t fi_group_gemm_nt(const void* A, const void* W, void* C,
                                       const int32_t* m_indptr, const int32_t* w_indices,
                                       int num_segments, int max_m_tiles, int N, int K,
                                       int64_t lda, int64_t ldw_n, int64_t ldw_seg,
                                       int64_t ldc, hipStream_t stream) {
  if (K % fi::ggemm::BK != 0) return hipErrorInvalidValue;
  dim3 grid((N + fi::ggemm::BN - 1) / fi::ggemm::BN, max_m_tiles, num_segments);
  hipLaunchKernelGGL(fi::ggemm::group_gemm_kernel, grid, dim3(fi::ggemm::NTH), 0, stream,
                     (const fi::bf16*)A, (const fi::bf16*)W, (fi::bf16*)C, m_indptr,
                     w_indices, num_segments, N, K, lda, ldw_n, ldw_seg, ldc);
  return hipGetLastError();
}
