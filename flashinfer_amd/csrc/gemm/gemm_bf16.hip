// bf16 NT GEMM on gfx950 MFMA (v_mfma_f32_32x32x16_bf16).
// C[M,N] = A[M,K] x B[N,K]^T  (both operands K-contiguous — the torch
// nn.Linear weight layout, and the layout every fragment load wants).
//
// Structure (guide §5 ladder, 128^2 tile, BK=64, 4 waves in 2x2, each wave a
// 64x64 output = 2x2 MFMA 32x32x16 subtiles; LDS tiles XOR-swizzled so
// ds_read_b128 is bank-conflict-free). Serves mm_bf16 (reference
// flashinfer/gemm/gemm_base.py:612) and validates the MFMA fragment layout
// contract in fi/frag.hpp for the attention kernels.
#include "fi/common.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"

namespace fi {

constexpr int BM = 128, BN = 128, BK = 64;

template <typename T>
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C, int M, int N,
    int K, int64_t lda, int64_t ldb, int64_t ldc, float alpha) {
  __shared__ T As[BM * BK];
  __shared__ T Bs[BN * BK];

  uint32_t nwg = gridDim.x * gridDim.y;
  uint32_t wg = xcd_swizzle(blockIdx.y * gridDim.x + blockIdx.x, nwg);
  int tiles_n = (N + BN - 1) / BN;
  int bm0 = (int)(wg / tiles_n) * BM;
  int bn0 = (int)(wg % tiles_n) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 64;  // wave row offset
  const int wn = (wid & 1) * 64;   // wave col offset
  const int line = lane & 31;
  const int khalf = (lane >> 5) * 8;

  floatx16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {};

  using frag = typename mfma_ab_frag<T>::type;

  for (int k0 = 0; k0 < K; k0 += BK) {
    // stage A and B tiles: 1024 16-byte units each over 256 threads
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int u = tid + it * 256;
      int row = u >> 3, chunk = u & 7;
      int gk = k0 + chunk * 8;
      // A
      {
        int gm = bm0 + row;
        shortx8 val = {};
        if (gm < M && gk < K)
          val = *reinterpret_cast<const shortx8*>(A + (int64_t)gm * lda + gk);
        *reinterpret_cast<shortx8*>(
            reinterpret_cast<char*>(As) + swz128(row * 128 + chunk * 16)) = val;
      }
      // B (N-major, K-contiguous)
      {
        int gn = bn0 + row;
        shortx8 val = {};
        if (gn < N && gk < K)
          val = *reinterpret_cast<const shortx8*>(B + (int64_t)gn * ldb + gk);
        *reinterpret_cast<shortx8*>(
            reinterpret_cast<char*>(Bs) + swz128(row * 128 + chunk * 16)) = val;
      }
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < BK / 16; ++ks) {
      frag a[2], b[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        a[i] = *reinterpret_cast<const frag*>(
            reinterpret_cast<const char*>(As) +
            swz128((wm + i * 32 + line) * 128 + (ks * 16 + khalf) * 2));
        b[i] = *reinterpret_cast<const frag*>(
            reinterpret_cast<const char*>(Bs) +
            swz128((wn + i * 32 + line) * 128 + (ks * 16 + khalf) * 2));
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = mfma_ab_frag<T>::mma32(a[i], b[j], acc[i][j]);
    }
    __syncthreads();
  }

  // epilogue: scalar stores (v1)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int m = bm0 + wm + i * 32 + mfma32_cd_row(r, lane);
        int n = bn0 + wn + j * 32 + mfma32_cd_col(lane);
        if (m < M && n < N) C[(int64_t)m * ldc + n] = from_f32<T>(acc[i][j][r] * alpha);
      }
    }
  }
}

}  // namespace fi

extern "C" hipError_t fi_gemm_nt(int dtype, const void* A, const void* B, void* C, int M,
                                 int N, int K, int64_t lda, int64_t ldb, int64_t ldc,
                                 float alpha, hipStream_t stream) {
  dim3 grid((N + fi::BN - 1) / fi::BN, (M + fi::BM - 1) / fi::BM), blk(256);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((fi::gemm_nt_kernel<fi::bf16>), grid, blk, 0, stream,
                         (const fi::bf16*)A, (const fi::bf16*)B, (fi::bf16*)C, M, N, K,
                         lda, ldb, ldc, alpha);
      break;
    case 1:
      hipLaunchKernelGGL((fi::gemm_nt_kernel<fi::fp16>), grid, blk, 0, stream,
                         (const fi::fp16*)A, (const fi::fp16*)B, (fi::fp16*)C, M, N, K,
                         lda, ldb, ldc, alpha);
      break;
    default:
      return hipErrorInvalidValue;
  }
  return hipGetLastError();
}
