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

// Index transform applied to selected local indices (reference
// flashinfer/topk.py top_k_page_table_transform:676 /
// top_k_ragged_transform:873 — DSA sparse-attention index translation):
//   mode 0: out = i
//   mode 1: out = i + offsets[row]
//   mode 2: out = page_table[batch, pt_start + i/page_size]*page_size + i%page_size
struct TopkTransform {
  const int32_t* offsets;               // mode 1
  const int32_t* page_table;            // mode 2 [B, pt_cols]
  const int32_t* row_to_batch;          // mode 2, optional
  const int32_t* page_table_row_starts; // mode 2, optional
  int pt_cols, page_size, mode;
};

__device__ __forceinline__ int32_t tk_transform(const TopkTransform& t, int row,
                                                int i) {
  if (t.mode == 1) return i + t.offsets[row];
  if (t.mode == 2) {
    int b = t.row_to_batch ? t.row_to_batch[row] : row;
    int ps = t.page_table_row_starts ? t.page_table_row_starts[row] : 0;
    int32_t page = t.page_table[(int64_t)b * t.pt_cols + ps + i / t.page_size];
    return page * t.page_size + i % t.page_size;
  }
  return i;
}

// one block per row; unordered top-k (values + indices). lengths/row_starts
// give a ragged window [row_starts[row], +lengths[row]) within the row;
// positions beyond min(k, len) pad as (-inf, -1). tie_break 1/2 bounds which
// boundary-value indices are selected (smaller / larger preferred) via a
// second binary search over the index.
__global__ void topk_kernel(const float* __restrict__ x, float* __restrict__ out_v,
                            int32_t* __restrict__ out_i, const int32_t* __restrict__ lengths,
                            const int32_t* __restrict__ row_starts, TopkTransform tr,
                            int rows, int d, int k, int64_t stride, int tie_break) {
  __shared__ float smem[TW + 1];
  __shared__ int s_cnt;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const int base = row_starts ? row_starts[row] : 0;
    const int len = lengths ? min(lengths[row], d - base) : d;
    const int keff = min(k, len);
    const float* xr = x + (int64_t)row * stride + base;
    // pad the tail first
    for (int p = keff + threadIdx.x; p < k; p += TB) {
      if (out_v) out_v[(int64_t)row * k + p] = -INFINITY;
      out_i[(int64_t)row * k + p] = -1;
    }
    if (keff == 0) { __syncthreads(); continue; }
    // threshold tau: largest value with |{v >= tau}| >= keff
    uint32_t lo = 0, hi = 0xFFFFFFFFu;
    for (int it = 0; it < 33 && lo < hi; ++it) {
      uint32_t mid = hi - ((hi - lo) >> 1);  // ceil midpoint, overflow-safe
      float tau = tk_u2f(mid);
      float cnt = 0.f;
      for (int i = threadIdx.x; i < len; i += TB)
        if (xr[i] >= tau) cnt += 1.f;
      float g = tk_block_sum(cnt, smem);
      if (g >= (float)keff) lo = mid;
      else hi = mid - 1;
    }
    float tau = tk_u2f(lo);
    // count strictly greater -> how many ties to take
    float cg = 0.f;
    for (int i = threadIdx.x; i < len; i += TB)
      if (xr[i] > tau) cg += 1.f;
    int n_greater = (int)tk_block_sum(cg, smem);
    int n_ties = keff - n_greater;
    // tie index bound: smallest B with |{i < B : x[i]==tau}| >= n_ties
    // (tie_break 1), or mirrored from the top (tie_break 2)
    int tie_lo = 0, tie_hi = len;  // selected ties: [tie_lo, tie_hi)
    if (tie_break == 1) {
      int blo = 0, bhi = len;
      while (blo < bhi) {
        int mid = blo + (bhi - blo) / 2;
        float c = 0.f;
        for (int i = threadIdx.x; i < mid; i += TB)
          if (xr[i] == tau) c += 1.f;
        if ((int)tk_block_sum(c, smem) >= n_ties) bhi = mid;
        else blo = mid + 1;
      }
      tie_hi = blo;
    } else if (tie_break == 2) {
      int blo = 0, bhi = len;
      while (blo < bhi) {
        int mid = blo + (bhi - blo + 1) / 2;
        float c = 0.f;
        for (int i = threadIdx.x + mid; i < len; i += TB)
          if (xr[i] == tau) c += 1.f;
        if ((int)tk_block_sum(c, smem) >= n_ties) blo = mid;
        else bhi = mid - 1;
      }
      tie_lo = blo;
    }
    // compact: strictly-greater first, then the bounded ties
    if (threadIdx.x == 0) s_cnt = 0;
    __syncthreads();
    for (int i = threadIdx.x; i < len; i += TB) {
      float v = xr[i];
      if (v > tau) {
        int pos = atomicAdd(&s_cnt, 1);
        if (pos < keff) {
          if (out_v) out_v[(int64_t)row * k + pos] = v;
          out_i[(int64_t)row * k + pos] = tk_transform(tr, row, i);
        }
      }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < len; i += TB) {
      float v = xr[i];
      if (v == tau && i >= tie_lo && i < tie_hi) {
        int pos = atomicAdd(&s_cnt, 1);
        if (pos < keff) {
          if (out_v) out_v[(int64_t)row * k + pos] = v;
          out_i[(int64_t)row * k + pos] = tk_transform(tr, row, i);
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

extern "C" hipError_t fi_topk(const float* x, float* out_v, int32_t* out_i,
                              const int32_t* lengths, const int32_t* row_starts,
                              const int32_t* offsets, const int32_t* page_table,
                              const int32_t* row_to_batch,
                              const int32_t* page_table_row_starts, int pt_cols,
                              int page_size, int mode, int rows, int d, int k,
                              int64_t stride, int tie_break, hipStream_t stream) {
  int grid = rows < 1024 ? rows : 1024;
  fi::TopkTransform tr{offsets, page_table, row_to_batch, page_table_row_starts,
                       pt_cols, page_size, mode};
  hipLaunchKernelGGL(fi::topk_kernel, dim3(grid), dim3(fi::TB), 0, stream, x, out_v,
                     out_i, lengths, row_starts, tr, rows, d, k, stride, tie_break);
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
