// Fused MoE routing / permute-build / quantizing data movers.
// Reference parity: csrc/fused_moe/trtllm_fused_moe_routing_deepseek.cu
// (no-aux-loss grouped top-k), trtllm_gen_routing topk-softmax,
// nv_internal moe_kernels.h:566 expand/permute stage. CDNA4 design:
//   * one WAVE per token for routing (E <= 256 scores live in 4 regs/lane
//     + a per-token LDS row for the grouped DSv3 selection);
//   * the permute build replaces the torch argsort/bincount/cumsum chain
//     with histogram -> tiny scan -> atomic scatter. Atomic order within an
//     expert is arbitrary but every downstream op is row-local (GEMM rows,
//     per-token-group quant) and the finalize gathers by the inverse map, so
//     outputs stay bitwise deterministic.
//   * gather+quant and silu_mul+quant fuse the fp8 activation quantization
//     into the data movers (one pass instead of write-then-requantize).
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int kEMax = 256;

// ---------------- top-k softmax routing (Mixtral-style) ----------------
// grid: ceil(T/4) x block 256 (4 waves, one token per wave)
__global__ void topk_softmax_kernel(const float* __restrict__ logits,
                                    float* __restrict__ weights,
                                    int32_t* __restrict__ ids, int T, int E,
                                    int k, int renorm) {
  int t = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (t >= T) return;
  int lane = threadIdx.x & 63;
  float s[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int e = lane + j * 64;
    s[j] = e < E ? logits[(int64_t)t * E + e] : -INFINITY;
  }
  float m = fmaxf(fmaxf(s[0], s[1]), fmaxf(s[2], s[3]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  float d = 0.f;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    s[j] = s[j] == -INFINITY ? 0.f : __builtin_expf(s[j] - m);
    d += s[j];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) d += __shfl_xor(d, off, 64);
  float inv_d = 1.f / d;
  float wsum = 0.f;
  for (int i = 0; i < k; ++i) {
    // wave argmax over the remaining probabilities
    float best = -1.f;
    int bj = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (s[j] > best) { best = s[j]; bj = j; }
    int be = lane + bj * 64;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ob = __shfl_xor(best, off, 64);
      int oe = __shfl_xor(be, off, 64);
      if (ob > best || (ob == best && oe < be)) { best = ob; be = oe; }
    }
    if (lane == 0) {
      weights[(int64_t)t * k + i] = best * inv_d;
      ids[(int64_t)t * k + i] = be;
    }
    wsum += best * inv_d;
    if (lane == (be & 63)) s[be >> 6] = -1.f;  // knock out the winner
  }
  if (renorm && lane == 0) {
    float r = 1.f / fmaxf(wsum, 1e-20f);
    for (int i = 0; i < k; ++i) weights[(int64_t)t * k + i] *= r;
  }
}

// ------------- DeepSeek-V3 no-aux-loss grouped top-k routing -------------
// sigmoid scores (+bias for SELECTION only); keep topk_group groups ranked
// by their top-2 sum; top-k over surviving experts; weights from the
// UNBIASED scores, normalized * routed_scaling_factor.
__global__ void dsv3_routing_kernel(const float* __restrict__ logits,
                                    const float* __restrict__ bias,
                                    float* __restrict__ weights,
                                    int32_t* __restrict__ ids, int T, int E,
                                    int k, int n_group, int topk_group,
                                    float scale) {
  __shared__ float sel_s[4][kEMax];
  __shared__ float gsc_s[4][32];
  int wv = threadIdx.x >> 6;
  int t = blockIdx.x * 4 + wv;
  if (t >= T) return;
  int lane = threadIdx.x & 63;
  int gsize = E / n_group;
  float sc[4], sel[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int e = lane + j * 64;
    if (e < E) {
      float v = 1.f / (1.f + __builtin_expf(-logits[(int64_t)t * E + e]));
      sc[j] = v;
      sel[j] = bias ? v + bias[e] : v;
    } else {
      sc[j] = 0.f;
      sel[j] = -INFINITY;
    }
    if (lane + j * 64 < E) sel_s[wv][lane + j * 64] = sel[j];
  }
  // per-group top-2 sum (lane g scans its group serially; gsize <= 32)
  if (lane < n_group) {
    float m1 = -INFINITY, m2 = -INFINITY;
    for (int i = 0; i < gsize; ++i) {
      float v = sel_s[wv][lane * gsize + i];
      if (v > m1) { m2 = m1; m1 = v; }
      else if (v > m2) m2 = v;
    }
    gsc_s[wv][lane] = m1 + m2;
  }
  // group top-k mask (lane 0, n_group <= 32): knock out losers
  if (lane == 0) {
    for (int drop = 0; drop < n_group - topk_group; ++drop) {
      float worst = INFINITY;
      int wg = -1;
      for (int g = 0; g < n_group; ++g) {
        if (gsc_s[wv][g] < worst) { worst = gsc_s[wv][g]; wg = g; }
      }
      gsc_s[wv][wg] = INFINITY;  // mark dropped
      for (int i = 0; i < gsize; ++i) sel_s[wv][wg * gsize + i] = -INFINITY;
    }
  }
  // re-read masked selection scores
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int e = lane + j * 64;
    sel[j] = e < E ? sel_s[wv][e] : -INFINITY;
  }
  // top-k by biased score; weight = unbiased score
  float wsum = 0.f;
  float myw[16];
  int mye[16];
  for (int i = 0; i < k; ++i) {
    float best = -INFINITY;
    int bj = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (sel[j] > best) { best = sel[j]; bj = j; }
    int be = lane + bj * 64;
    float bw = sc[bj];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ob = __shfl_xor(best, off, 64);
      int oe = __shfl_xor(be, off, 64);
      float ow = __shfl_xor(bw, off, 64);
      if (ob > best || (ob == best && oe < be)) { best = ob; be = oe; bw = ow; }
    }
    myw[i] = bw;
    mye[i] = be;
    wsum += bw;
    if (lane == (be & 63)) sel[be >> 6] = -INFINITY;
  }
  if (lane == 0) {
    float r = scale / fmaxf(wsum, 1e-20f);
    for (int i = 0; i < k; ++i) {
      weights[(int64_t)t * k + i] = myw[i] * r;
      ids[(int64_t)t * k + i] = mye[i];
    }
  }
}

// ---------------- permute build: hist -> scan -> scatter ----------------
__global__ void moe_hist_kernel(const int32_t* __restrict__ ids,
                                int32_t* __restrict__ counts, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) atomicAdd(counts + ids[i], 1);
}

// one workgroup: exclusive scan of counts into m_indptr (segment starts
// 128-aligned so every GEMM M-tile belongs to exactly ONE expert — the
// grouped GEMM runs a flat tile grid with a per-WG segment binary search
// instead of a z-dim padded to the worst-case tile count), zero the cursors
__global__ void moe_scan_kernel(const int32_t* __restrict__ counts,
                                int32_t* __restrict__ m_indptr,
                                int32_t* __restrict__ cursor, int E, int align) {
  if (threadIdx.x == 0) {
    int acc = 0;
    m_indptr[0] = 0;
    for (int e = 0; e < E; ++e) {
      acc += counts[e];
      if (align > 1) acc = (acc + align - 1) / align * align;
      m_indptr[e + 1] = acc;
    }
  }
  for (int e = threadIdx.x; e < E; e += blockDim.x) cursor[e] = 0;
}

__global__ void moe_scatter_kernel(const int32_t* __restrict__ ids,
                                   const int32_t* __restrict__ m_indptr,
                                   int32_t* __restrict__ cursor,
                                   int32_t* __restrict__ token_of_copy,
                                   int32_t* __restrict__ inv, int n, int k) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int e = ids[i];
  int pos = m_indptr[e] + atomicAdd(cursor + e, 1);
  token_of_copy[pos] = i / k;
  inv[i] = pos;
}

// ------------- fused gather + per-128-group fp8 quantization -------------
// dst_q[r] = e4m3(src[token_of_copy[r]] / s); scales: f32 [K/128, R]
// (MN-major) or — E8M0 mode — u8 e8m0 bytes [R, K/128] (scale rounded UP to
// a power of two so the MX MFMA applies it in hardware).
// block 256 = 4 waves; wave handles one row's groups strided by 4.

// ---- vectorized per-128-group fp8 quant store: a WAVE covers 4 groups of
// 128 (lane j holds 8 contiguous f32 values = 16B loads), the group amax is
// a segmented 16-lane shfl tree, and the 8 fp8 bytes store as one u64.
// (The first version loaded 2 elements per lane -> 2.55 TB/s measured on
// the MoE loop; this one is load/store-width bound.)
template <bool E8M0>
__device__ __forceinline__ void quant_group_store(
    const float (&v)[8], int lane, int64_t r, int g, int R, int nG,
    uint8_t* __restrict__ dstrow, float* __restrict__ scale, int base) {
  float amax = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(v[j]));
#pragma unroll
  for (int off = 8; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 16));
  float inv_s;
  if constexpr (E8M0) {
    int e;
    frexpf(fmaxf(amax, 1e-10f) / 448.f, &e);
    inv_s = ldexpf(1.f, -e);
    if ((lane & 15) == 0)
      reinterpret_cast<uint8_t*>(scale)[r * nG + g] = (uint8_t)(127 + e);
  } else {
    float sv = fmaxf(amax, 1e-10f) / 448.f;
    inv_s = 1.f / sv;
    if ((lane & 15) == 0) scale[(int64_t)g * R + r] = sv;
  }
  uint64_t packed = 0;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    packed |= (uint64_t)__builtin_bit_cast(uint8_t,
                                           from_f32<fp8_e4m3>(v[j] * inv_s))
              << (8 * j);
  *reinterpret_cast<uint64_t*>(dstrow + base) = packed;
}

template <typename T, bool E8M0 = false>
__global__ void gather_quant_kernel(const T* __restrict__ src,
                                    const int32_t* __restrict__ token_of_copy,
                                    uint8_t* __restrict__ dst,
                                    float* __restrict__ scale, int R, int K) {
  int r = blockIdx.x;
  if (r >= R) return;
  const T* row = src + (int64_t)token_of_copy[r] * K;
  uint8_t* drow = dst + (int64_t)r * K;
  int lane = threadIdx.x & 63;
  int wv = threadIdx.x >> 6;
  const int nG = K / 128;
  // 512-elem wave clusters (4 groups); K % 512 tail handled per-group below
  for (int c = wv; c < K / 512; c += 4) {
    int base = c * 512 + lane * 8;
    vec_t<T, 8> xv;
    xv.load(row + base);
    float v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = xv.get(j);
    quant_group_store<E8M0>(v, lane, r, c * 4 + (lane >> 4), R, nG, drow,
                            scale, base);
  }
  for (int g = (K / 512) * 4 + wv; g < nG; g += 4) {
    // tail groups: 16 lanes each would underuse the wave — keep all 64
    // lanes on one group, 2 elems per lane (rare: K % 512 != 0)
    float v0 = to_f32<T>(row[g * 128 + lane * 2]);
    float v1 = to_f32<T>(row[g * 128 + lane * 2 + 1]);
    float amax = fmaxf(fabsf(v0), fabsf(v1));
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, 64));
    float inv_s;
    if constexpr (E8M0) {
      int e;
      frexpf(fmaxf(amax, 1e-10f) / 448.f, &e);
      inv_s = ldexpf(1.f, -e);
      if (lane == 0)
        reinterpret_cast<uint8_t*>(scale)[(int64_t)r * nG + g] =
            (uint8_t)(127 + e);
    } else {
      float sv = fmaxf(amax, 1e-10f) / 448.f;
      inv_s = 1.f / sv;
      if (lane == 0) scale[(int64_t)g * R + r] = sv;
    }
    uint16_t pair =
        (uint16_t)__builtin_bit_cast(uint8_t, from_f32<fp8_e4m3>(v0 * inv_s)) |
        ((uint16_t)__builtin_bit_cast(uint8_t, from_f32<fp8_e4m3>(v1 * inv_s))
         << 8);
    reinterpret_cast<uint16_t*>(drow)[g * 64 + lane] = pair;
  }
}

// ---------- fused silu(gate)*up + per-128-group fp8 quantization ----------
// h [R, 2I] (gate | up) -> q [R, I] e4m3 + scales [I/128, R]
template <typename T, bool E8M0 = false>
__global__ void silu_mul_quant_kernel(const T* __restrict__ h,
                                      uint8_t* __restrict__ dst,
                                      float* __restrict__ scale, int R, int I,
                                      int gelu) {
  int r = blockIdx.x;
  if (r >= R) return;
  const T* gate = h + (int64_t)r * (2 * I);
  const T* up = gate + I;
  uint8_t* drow = dst + (int64_t)r * I;
  int lane = threadIdx.x & 63;
  int wv = threadIdx.x >> 6;
  const int nG = I / 128;
  auto act_mul = [&](float x, float u) {
    float act;
    if (gelu) {
      act = 0.5f * x * (1.f + tanhf(0.7978845608028654f *
                                    (x + 0.044715f * x * x * x)));
    } else {
      act = x / (1.f + __builtin_expf(-x));
    }
    return act * u;
  };
  for (int c = wv; c < I / 512; c += 4) {
    int base = c * 512 + lane * 8;
    vec_t<T, 8> gv, uv;
    gv.load(gate + base);
    uv.load(up + base);
    float v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = act_mul(gv.get(j), uv.get(j));
    quant_group_store<E8M0>(v, lane, r, c * 4 + (lane >> 4), R, nG, drow,
                            scale, base);
  }
  for (int g = (I / 512) * 4 + wv; g < nG; g += 4) {
    float v0 = act_mul(to_f32<T>(gate[g * 128 + lane * 2]),
                       to_f32<T>(up[g * 128 + lane * 2]));
    float v1 = act_mul(to_f32<T>(gate[g * 128 + lane * 2 + 1]),
                       to_f32<T>(up[g * 128 + lane * 2 + 1]));
    float amax = fmaxf(fabsf(v0), fabsf(v1));
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, 64));
    float inv_s;
    if constexpr (E8M0) {
      int e;
      frexpf(fmaxf(amax, 1e-10f) / 448.f, &e);
      inv_s = ldexpf(1.f, -e);
      if (lane == 0)
        reinterpret_cast<uint8_t*>(scale)[(int64_t)r * nG + g] =
            (uint8_t)(127 + e);
    } else {
      float sv = fmaxf(amax, 1e-10f) / 448.f;
      inv_s = 1.f / sv;
      if (lane == 0) scale[(int64_t)g * R + r] = sv;
    }
    uint16_t pair =
        (uint16_t)__builtin_bit_cast(uint8_t, from_f32<fp8_e4m3>(v0 * inv_s)) |
        ((uint16_t)__builtin_bit_cast(uint8_t, from_f32<fp8_e4m3>(v1 * inv_s))
         << 8);
    reinterpret_cast<uint16_t*>(drow)[g * 64 + lane] = pair;
  }
}

}  // namespace fi

extern "C" {

hipError_t fi_moe_topk_softmax(const float* logits, float* weights, int32_t* ids,
                               int T, int E, int k, int renorm,
                               hipStream_t stream) {
  if (E > fi::kEMax || k > 16) return hipErrorInvalidValue;
  hipLaunchKernelGGL(fi::topk_softmax_kernel, dim3((T + 3) / 4), dim3(256), 0,
                     stream, logits, weights, ids, T, E, k, renorm);
  return hipGetLastError();
}

hipError_t fi_dsv3_routing(const float* logits, const float* bias, float* weights,
                           int32_t* ids, int T, int E, int k, int n_group,
                           int topk_group, float scale, hipStream_t stream) {
  if (E > fi::kEMax || k > 16 || n_group > 32 || E / n_group > 32)
    return hipErrorInvalidValue;
  hipLaunchKernelGGL(fi::dsv3_routing_kernel, dim3((T + 3) / 4), dim3(256), 0,
                     stream, logits, bias, weights, ids, T, E, k, n_group,
                     topk_group, scale);
  return hipGetLastError();
}

hipError_t fi_moe_build_permute(const int32_t* ids, int32_t* counts,
                                int32_t* m_indptr, int32_t* cursor,
                                int32_t* token_of_copy, int32_t* inv, int n,
                                int k, int E, int align, hipStream_t stream) {
  hipLaunchKernelGGL(fi::moe_hist_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     stream, ids, counts, n);
  hipLaunchKernelGGL(fi::moe_scan_kernel, dim3(1), dim3(256), 0, stream, counts,
                     m_indptr, cursor, E, align);
  hipLaunchKernelGGL(fi::moe_scatter_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     stream, ids, m_indptr, cursor, token_of_copy, inv, n, k);
  return hipGetLastError();
}

hipError_t fi_gather_quant(int dtype, const void* src, const int32_t* token_of_copy,
                           uint8_t* dst, float* scale, int R, int K, int e8m0,
                           hipStream_t stream) {
  if (K % 128) return hipErrorInvalidValue;
#define LAUNCH_GQ(T)                                                          \
  do {                                                                        \
    if (e8m0)                                                                 \
      hipLaunchKernelGGL((fi::gather_quant_kernel<T, true>), dim3(R),         \
                         dim3(256), 0, stream, (const T*)src, token_of_copy,  \
                         dst, scale, R, K);                                   \
    else                                                                      \
      hipLaunchKernelGGL((fi::gather_quant_kernel<T, false>), dim3(R),        \
                         dim3(256), 0, stream, (const T*)src, token_of_copy,  \
                         dst, scale, R, K);                                   \
  } while (0)
  switch (dtype) {
    case 0: LAUNCH_GQ(fi::bf16); break;
    case 1: LAUNCH_GQ(fi::fp16); break;
    case 2: LAUNCH_GQ(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH_GQ
  return hipGetLastError();
}

hipError_t fi_silu_mul_quant(int dtype, const void* h, uint8_t* dst, float* scale,
                             int R, int I, int gelu, int e8m0,
                             hipStream_t stream) {
  if (I % 128) return hipErrorInvalidValue;
#define LAUNCH_SQ(T)                                                            \
  do {                                                                          \
    if (e8m0)                                                                   \
      hipLaunchKernelGGL((fi::silu_mul_quant_kernel<T, true>), dim3(R),         \
                         dim3(256), 0, stream, (const T*)h, dst, scale, R, I,   \
                         gelu);                                                 \
    else                                                                        \
      hipLaunchKernelGGL((fi::silu_mul_quant_kernel<T, false>), dim3(R),        \
                         dim3(256), 0, stream, (const T*)h, dst, scale, R, I,   \
                         gelu);                                                 \
  } while (0)
  switch (dtype) {
    case 0: LAUNCH_SQ(fi::bf16); break;
    case 1: LAUNCH_SQ(fi::fp16); break;
    case 2: LAUNCH_SQ(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH_SQ
  return hipGetLastError();
}

}  // extern "C"
