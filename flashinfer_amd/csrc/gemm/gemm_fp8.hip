// FP8 (e4m3) groupwise-scaled NT GEMM for gfx950 — dense, grouped (MoE /
// segment) and batched in one kernel. Parity targets: reference
// gemm_fp8_nt_groupwise (1x128 activation scales x 128x128 weight-block
// scales, gemm/gemm_base.py:7589), group_gemm_fp8_nt_groupwise (:8103),
// bmm_fp8 (:7387, per-tensor scales).
//
// Design: gemm_v2 pipeline shape at BK = 128 fp8 elements (= one K scale
// group = 128-byte LDS rows, byte-identical swizzle geometry to the bf16
// BK=64 kernels), v_mfma_f32_32x32x16_fp8_fp8 (8-byte fragments), per-group
// f32 rescale of the local accumulator into the master accumulator:
//   C += a_scale[kg, m] * b_scale[kg, n/128] * sum_k A_q B_q.
// Per-tensor mode (bmm_fp8): scales null, one scalar factor at the end.
#include "fi/common.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"

namespace fi {

namespace f8gemm {

constexpr int BM = 128, BN = 128, BK = 128;  // BK in fp8 elements (128 B rows)
// 8 waves of 64x32 output each (2x4 wave grid): the groupwise variant keeps
// TWO accumulators (per-K-group local + scaled master) — small per-wave
// tiles keep the pair inside the register budget (no spill).
constexpr int NTH = 512;
constexpr int WM = 64, WN = 32;

__device__ __forceinline__ void stage_tile(const uint8_t* __restrict__ gbase,
                                           int64_t ld, int row0, int rows_end, int k0,
                                           uint32_t lds_base_bytes, int tid) {
  // BM x BK x 1B = 16 KB = 1024 16-B units over 512 threads
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int u = tid + i * NTH;
    uint32_t dst_byte = (uint32_t)u * 16;
    // swz256 over ROW PAIRS (two 128-B rows share a 256-B bank row): a
    // 128-B-row swz128 leaves rows r/r+8 aliased -> 4-way conflicts on
    // 32-row fragment reads (PMC: LDSBankConflict 17.7); the 4-bit spread
    // de-conflicts all 32 rows. XOR is an involution, so pre-swizzling the
    // GLOBAL address keeps the DMA destination linear (guide s09 pattern).
    uint32_t logical = swz256(dst_byte);
    int row = logical >> 7;
    int col = logical & 127;
    int gm = row0 + row;
    const uint8_t* src =
        gbase + (int64_t)(gm < rows_end ? gm : rows_end - 1) * ld + k0 + col;
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)src,
                                     (__attribute__((address_space(3))) void*)(uintptr_t)(
                                         lds_base_bytes + dst_byte),
                                     16, 0, 0);
  }
}

// MODE: 0 = per-tensor scalar scale; 1 = f32 groupwise (1x128 activation x
// 128x128 weight-block scales, local-accumulator rescale per K group);
// 2 = MX e8m0 (hardware-applied per-row scale bytes in the f8f6f4
// instruction: per-lane activation byte + wave-uniform weight-block byte —
// scale semantics probed in scripts/probe/mfma_mx5/6/7.hip). A 64x64-wave
// variant and a triple-buffer variant were both tried and measured SLOWER
// (co-residency loss; profiles/README r02) — this 8-wave 32x64 shape stays.
template <int MODE>
__global__ __launch_bounds__(NTH, 2) void gemm_fp8_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ W,
    bf16* __restrict__ C, const int32_t* __restrict__ m_indptr,
    const int32_t* __restrict__ w_indices, int N, int K, int64_t lda, int64_t ldw_n,
    int64_t ldw_seg, int64_t ldc, const float* __restrict__ a_scales,  // [K/128, Mtot]
    const float* __restrict__ b_scales,  // [S, K/128, N/128]
    float scalar_scale, int64_t a_scale_stride, int flat_segments) {
  __shared__ uint8_t As[2][BM * BK];
  __shared__ uint8_t Bs[2][BN * BK];
  __shared__ float ascale_s[2][BM];  // per-row activation scales for this kt

  // FLAT grid (num_segments passed > 0): blockIdx.y is a GLOBAL M tile over
  // 128-aligned segments; binary-search the owning segment. Legacy grid
  // (num_segments == 0): blockIdx.z is the segment, y the tile within —
  // kept for unaligned segment GEMM callers (LoRA / plain groupwise).
  // XCD-aware block swizzle (guide T1): the dispatcher round-robins
  // linear block ids across the 8 XCD L2s, so consecutive ids — which
  // share an A panel (same m-tile, neighboring n-tiles) — would each
  // refetch A from HBM. Remapping gives each XCD a CONTIGUOUS run of the
  // grid, so the shared A panel stays in that XCD's L2. Bijective only
  // when nwg % 8 == 0 (guide errata #11) — identity otherwise.
  int bx = blockIdx.x, by = blockIdx.y;
  if (gridDim.z == 1) {
    const int nwg = gridDim.x * gridDim.y;
    if ((nwg & 7) == 0) {
      int bid = bx + gridDim.x * by;
      int sw = (bid & 7) * (nwg >> 3) + (bid >> 3);
      bx = sw % gridDim.x;
      by = sw / gridDim.x;
    }
  }
  int seg, m0, m_end;
  if (flat_segments > 0) {
    m0 = by * BM;
    int lo = 0, hi = flat_segments - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (m_indptr[mid] <= m0) lo = mid;
      else hi = mid - 1;
    }
    seg = lo;
    m_end = m_indptr[seg + 1];
  } else {
    seg = blockIdx.z;
    m0 = m_indptr[seg] + by * BM;
    m_end = m_indptr[seg + 1];
  }
  if (m0 >= m_end) return;
  const int bn0 = bx * BN;
  const int widx = w_indices ? w_indices[seg] : seg;
  const uint8_t* Wb = W + (int64_t)widx * ldw_seg;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  // C^T wave tiling (32 m x 64 n per wave): the accumulator's LANE COLUMN is
  // the M row, so the per-row activation scale is a per-lane SCALAR — the
  // straight-C layout needed 32 ds_read_b32 of ascale_s per tile per wave
  // (the DS-pipe hotspot at 839-919 TF; profiles/README r02 fp8 entry).
  const int wm = (wid & 3) * 32;
  const int wn = (wid >> 2) * 64;
  const int line = lane & 31;

  floatx16 accm[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) accm[i] = {};

  const uint32_t as_base = (uint32_t)(uintptr_t)&As[0][0];
  const uint32_t bs_base = (uint32_t)(uintptr_t)&Bs[0][0];
  constexpr uint32_t BUF_BYTES = BM * BK;

  int nk = K / BK;
  stage_tile(A, lda, m0, m_end, 0, as_base, tid);
  stage_tile(Wb, ldw_n, bn0, N, 0, bs_base, tid);
  if (MODE == 1 && tid < BM) {
    int m = m0 + tid;
    ascale_s[0][tid] = a_scales[m < m_end ? m : m_end - 1];
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < nk; ++kt) {
    if (kt + 1 < nk) {
      stage_tile(A, lda, m0, m_end, (kt + 1) * BK, as_base + (cur ^ 1) * BUF_BYTES, tid);
      stage_tile(Wb, ldw_n, bn0, N, (kt + 1) * BK, bs_base + (cur ^ 1) * BUF_BYTES, tid);
      if (MODE == 1 && tid < BM) {
        int m = m0 + tid;
        ascale_s[cur ^ 1][tid] =
            a_scales[(int64_t)(kt + 1) * a_scale_stride + (m < m_end ? m : m_end - 1)];
      }
    }
    const char* a_lds = (const char*)&As[cur][0];
    const char* b_lds = (const char*)&Bs[cur][0];
    floatx16 acc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[i] = {};
    // 32-byte lane fragments for the large-K (x64) fp8 MFMA: two 16-B LDS
    // reads (the swizzle permutes 16-B blocks within a 128-B row)
    union frag32 {
      intx8 v;
      uint4 q[2];
    };
    auto ld32 = [&](const char* base, int row, int kb) {
      frag32 u;
      u.q[0] = *reinterpret_cast<const uint4*>(base + swz256(row * BK + kb));
      u.q[1] = *reinterpret_cast<const uint4*>(base + swz256(row * BK + kb + 16));
      return u.v;
    };
    const int kh32 = (lane >> 5) * 32;
    int vb = 127, wsb = 127;
    if constexpr (MODE == 2) {
      // activation scale byte for this lane's M row + this K group; both
      // lane halves carry the same byte (HW averages 16-elem sub-blocks)
      int m = m0 + wm + line;
      vb = ((const uint8_t*)a_scales)[(int64_t)(m < m_end ? m : m_end - 1) *
                                          (K / BK) + kt];
      wsb = ((const uint8_t*)b_scales)[((int64_t)widx * (K / BK) + kt) *
                                           ((N + 127) / 128) + bn0 / 128];
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int g2 = 0; g2 < 2; ++g2) {  // two K=64 groups per BK=128 tile
      int kb = g2 * 64 + kh32;
      intx8 afrag = ld32(a_lds, wm + line, kb);  // B-operand: col = m
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        if constexpr (MODE == 2) {
          accm[i] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              ld32(b_lds, wn + i * 32 + line, kb), afrag, accm[i], 0, 0, 0,
              wsb, 0, vb);
        } else {
          acc[i] = mfma_32x32x64_fp8(ld32(b_lds, wn + i * 32 + line, kb), afrag,
                                     acc[i]);
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
    // rescale local accumulator into master (per-lane scalar sa*sb)
    if constexpr (MODE == 1) {
      float sb = b_scales[((int64_t)seg * nk + kt) * ((N + 127) / 128) + bn0 / 128];
      float sab = ascale_s[cur][wm + line] * sb;
      // packed f32 FMA halves the rescale VALU chain (v_pk_fma_f32)
      typedef __attribute__((ext_vector_type(2))) float f32x2;
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        f32x2* am = reinterpret_cast<f32x2*>(&accm[i]);
        const f32x2* ac = reinterpret_cast<const f32x2*>(&acc[i]);
#pragma unroll
        for (int r = 0; r < 8; ++r) am[r] += ac[r] * sab;
      }
    } else if constexpr (MODE == 0) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int r = 0; r < 16; ++r) accm[i][r] += acc[i][r];
    }
    // Full vmcnt drain before the barrier is REQUIRED here: the prefetch
    // distance is one tile (double-buffered LDS), so the tile read next
    // iteration is the one whose DMAs are still in flight at this barrier —
    // a vmcnt(4) partial wait was tried (+46 TF) and raced (caught by the
    // 64-expert MoE e2e test: nondeterministic 8-column stripes). Deeper
    // pipelining needs a third LDS buffer, not a looser wait.
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  float fs = MODE == 0 ? scalar_scale : 1.f;
  // C^T epilogue: lane column = m, accumulator rows = n
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int m = m0 + wm + mfma32_cd_col(lane);
      int n = bn0 + wn + i * 32 + mfma32_cd_row(r, lane);
      if (m < m_end && n < N)
        C[(int64_t)m * ldc + n] = from_f32<bf16>(accm[i][r] * fs);
    }
  }
}

}  // namespace f8gemm

}  // namespace fi

extern "C" hipError_t fi_gemm_fp8_grouped(
    const void* A, const void* W, void* C, const int32_t* m_indptr,
    const int32_t* w_indices, int num_segments, int max_m_tiles, int N, int K,
    int64_t lda, int64_t ldw_n, int64_t ldw_seg, int64_t ldc, const float* a_scales,
    const float* b_scales, float scalar_scale, int64_t a_scale_stride,
    int flat_tiles, int mx, hipStream_t stream) {
  if (K % 128 != 0) return hipErrorInvalidValue;
  // flat < 0 => legacy z-grid; flat >= 0 ignored sentinel handled by caller
  dim3 grid, blk(fi::f8gemm::NTH);
  int flat_segs = 0;
  if (flat_tiles > 0) {
    grid = dim3((N + fi::f8gemm::BN - 1) / fi::f8gemm::BN, flat_tiles, 1);
    flat_segs = num_segments;
  } else {
    grid = dim3((N + fi::f8gemm::BN - 1) / fi::f8gemm::BN, max_m_tiles,
                num_segments);
  }
  int mode = (a_scales && b_scales) ? (mx ? 2 : 1) : 0;
#define LG(M)                                                                     \
  hipLaunchKernelGGL((fi::f8gemm::gemm_fp8_kernel<M>), grid, blk, 0, stream,      \
                     (const uint8_t*)A, (const uint8_t*)W, (fi::bf16*)C, m_indptr,\
                     w_indices, N, K, lda, ldw_n, ldw_seg, ldc, a_scales,         \
                     b_scales, scalar_scale, a_scale_stride, flat_segs)
  if (mode == 2) LG(2);
  else if (mode == 1) LG(1);
  else LG(0);
#undef LG
  return hipGetLastError();
}
