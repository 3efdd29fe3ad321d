// mHC (multi-head hyper-connections) fused ops for gfx950. Reference parity:
// flashinfer/mhc.py mhc_post:76, mhc_pre_big_fuse:176,
// mhc_pre_big_fuse_with_prenorm:334 — math per the reference's own
// tests/mhc/test_mhc_pre_big_fuse.py reference implementation:
//   rstd   = rsqrt(sum_splits(sqrsum)/k + rms_eps)
//   mixes  = sum_splits(dot_mix) * rstd          (24 = pre 4 | post 4 | comb 16)
//   pre    = sigmoid(mixes[:4]*scale0 + base[:4]) + pre_eps
//   post   = sigmoid(mixes[4:8]*scale1 + base[4:8]) * post_mult
//   comb   = sinkhorn(softmax(mixes[8:]*scale2 + base[8:]))   (HC x HC, R iters)
//   layer_input[h] = sum_hc pre[hc] * residual[hc, h]
// and mhc_post: out[new,h] = x[h]*post[new] + sum_old residual[old,h]*comb[old,new].
// One 256-thread block per token: the 4x4 scalar math runs on thread 0 and is
// broadcast through LDS; the H-wide mixing streams bf16x8 vectors.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int kHC = 4;

// out [T, 4, H] = x [T, H] * post[T, 4] + residual [T, 4, H] @ comb [T, 4, 4]
template <typename T, int VEC>
__global__ void mhc_post_kernel(const T* __restrict__ x, const T* __restrict__ residual,
                                const float* __restrict__ post_mix,
                                const float* __restrict__ comb_mix, T* __restrict__ out,
                                int64_t tokens, int H) {
  for (int64_t tok = blockIdx.x; tok < tokens; tok += gridDim.x) {
    const T* xr = x + tok * H;
    const T* rr = residual + tok * kHC * H;
    T* orow = out + tok * kHC * H;
    float post[kHC], comb[kHC][kHC];
#pragma unroll
    for (int n = 0; n < kHC; ++n) {
      post[n] = post_mix[tok * kHC + n];
#pragma unroll
      for (int o = 0; o < kHC; ++o) comb[o][n] = comb_mix[(tok * kHC + o) * kHC + n];
    }
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      vec_t<T, VEC> vx, vr[kHC], vo;
      vx.load(xr + i);
#pragma unroll
      for (int o = 0; o < kHC; ++o) vr[o].load(rr + (int64_t)o * H + i);
#pragma unroll
      for (int n = 0; n < kHC; ++n) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float v = vx.get(j) * post[n];
#pragma unroll
          for (int o = 0; o < kHC; ++o) v += vr[o].get(j) * comb[o][n];
          vo.set(j, v);
        }
        vo.store(orow + (int64_t)n * H + i);
      }
    }
  }
}

// PRENORM: sqrsum computed in-kernel over residual[tok] (flattened 4*H).
template <typename T, int VEC, bool PRENORM>
__global__ void mhc_pre_kernel(const float* __restrict__ dot_mix,
                               const float* __restrict__ sqrsum,
                               const T* __restrict__ residual,
                               const float* __restrict__ scale,   // [3]
                               const float* __restrict__ base,    // [24]
                               float* __restrict__ post_mix,      // [T, 4, 1]
                               float* __restrict__ comb_mix,      // [T, 4, 4]
                               T* __restrict__ layer_input,       // [T, H]
                               int64_t tokens, int H, int num_splits, float inv_k,
                               float rms_eps, float pre_eps, float sink_eps,
                               float post_mult, int sink_repeat) {
  __shared__ float s_red[4];
  __shared__ float s_pre[kHC];
  constexpr int ML = 2 * kHC + kHC * kHC;  // 24 logit slots
  for (int64_t tok = blockIdx.x; tok < tokens; tok += gridDim.x) {
    const T* rr = residual + tok * kHC * H;
    float ss = 0.f;
    if constexpr (PRENORM) {
      float part = 0.f;
      for (int i = threadIdx.x * VEC; i < kHC * H; i += blockDim.x * VEC) {
        vec_t<T, VEC> v;
        v.load(rr + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = v.get(j);
          part += f * f;
        }
      }
      part = wave_reduce_sum<kWaveSize>(part);
      int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
      if (l == 0) s_red[w] = part;
      __syncthreads();
      ss = s_red[0] + s_red[1] + s_red[2] + s_red[3];
    }
    if (threadIdx.x == 0) {
      if constexpr (!PRENORM) {
        ss = 0.f;
        for (int s = 0; s < num_splits; ++s) ss += sqrsum[s * tokens + tok];
      }
      float rstd = rsqrtf(ss * inv_k + rms_eps);
      float m[ML];
#pragma unroll
      for (int i = 0; i < ML; ++i) {
        float d = 0.f;
        for (int s = 0; s < num_splits; ++s)
          d += dot_mix[(s * tokens + tok) * ML + i];
        m[i] = d * rstd;
      }
#pragma unroll
      for (int i = 0; i < kHC; ++i) {
        float pre = 1.f / (1.f + __builtin_expf(-(m[i] * scale[0] + base[i])));
        s_pre[i] = pre + pre_eps;
        float post = 1.f / (1.f + __builtin_expf(-(m[kHC + i] * scale[1] + base[kHC + i])));
        post_mix[(tok * kHC + i)] = post * post_mult;
      }
      // comb: row-softmax then sinkhorn alternate row/col normalization
      float c[kHC][kHC];
#pragma unroll
      for (int r = 0; r < kHC; ++r) {
        float mx = -INFINITY;
#pragma unroll
        for (int q = 0; q < kHC; ++q) {
          c[r][q] = m[2 * kHC + r * kHC + q] * scale[2] + base[2 * kHC + r * kHC + q];
          mx = fmaxf(mx, c[r][q]);
        }
        float sum = 0.f;
#pragma unroll
        for (int q = 0; q < kHC; ++q) {
          c[r][q] = __builtin_expf(c[r][q] - mx);
          sum += c[r][q];
        }
#pragma unroll
        for (int q = 0; q < kHC; ++q) c[r][q] = c[r][q] / sum + sink_eps;
      }
      for (int it = 0; it < sink_repeat; ++it) {
        if (it > 0) {
#pragma unroll
          for (int r = 0; r < kHC; ++r) {
            float s2 = c[r][0] + c[r][1] + c[r][2] + c[r][3] + sink_eps;
#pragma unroll
            for (int q = 0; q < kHC; ++q) c[r][q] /= s2;
          }
        }
#pragma unroll
        for (int q = 0; q < kHC; ++q) {
          float s2 = c[0][q] + c[1][q] + c[2][q] + c[3][q] + sink_eps;
#pragma unroll
          for (int r = 0; r < kHC; ++r) c[r][q] /= s2;
        }
      }
#pragma unroll
      for (int r = 0; r < kHC; ++r)
#pragma unroll
        for (int q = 0; q < kHC; ++q) comb_mix[(tok * kHC + r) * kHC + q] = c[r][q];
    }
    __syncthreads();
    float pre[kHC] = {s_pre[0], s_pre[1], s_pre[2], s_pre[3]};
    T* lrow = layer_input + tok * H;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      vec_t<T, VEC> vr[kHC], vo;
#pragma unroll
      for (int o = 0; o < kHC; ++o) vr[o].load(rr + (int64_t)o * H + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float v = 0.f;
#pragma unroll
        for (int o = 0; o < kHC; ++o) v += pre[o] * vr[o].get(j);
        vo.set(j, v);
      }
      vo.store(lrow + i);
    }
    __syncthreads();
  }
}

// concat_mla_k (reference flashinfer/concat_ops.py concat_mla_k:32):
// k [T, Hk, nope+rope] <- [k_nope [T, Hk, nope] | k_rope [T, 1, rope] bcast]
template <typename T, int VEC>
__global__ void concat_mla_k_kernel(T* __restrict__ k, const T* __restrict__ k_nope,
                                    const T* __restrict__ k_rope, int64_t tokens,
                                    int Hk, int nope, int rope) {
  const int D = nope + rope;
  for (int64_t tok = blockIdx.x; tok < tokens; tok += gridDim.x) {
    const T* nrow = k_nope + tok * Hk * nope;
    const T* rrow = k_rope + tok * rope;
    T* krow = k + tok * Hk * D;
    for (int e = threadIdx.x * VEC; e < Hk * D; e += blockDim.x * VEC) {
      int h = e / D, d = e % D;
      vec_t<T, VEC> v;
      if (d + VEC <= nope) v.load(nrow + h * nope + d);
      else if (d >= nope) v.load(rrow + d - nope);
      else {  // straddles the boundary (only when nope % VEC != 0)
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          int dj = d + j;
          v.data[j] = dj < nope ? nrow[h * nope + dj] : rrow[dj - nope];
        }
      }
      v.store(krow + e);
    }
  }
}

}  // namespace fi

extern "C" hipError_t fi_mhc_post(int dtype, const void* x, const void* residual,
                                  const float* post_mix, const float* comb_mix,
                                  void* out, int64_t tokens, int H,
                                  hipStream_t stream) {
  int grid = tokens < 4096 ? (int)tokens : 4096;
  if (grid == 0) return hipSuccess;
  dim3 g(grid), blk(256);
#define LMP(T, V)                                                              \
  hipLaunchKernelGGL((fi::mhc_post_kernel<T, V>), g, blk, 0, stream, (const T*)x, \
                     (const T*)residual, post_mix, comb_mix, (T*)out, tokens, H)
  bool v8 = (H % 8 == 0);
  switch (dtype) {
    case 0: if (v8) LMP(fi::bf16, 8); else LMP(fi::bf16, 1); break;
    case 1: if (v8) LMP(fi::fp16, 8); else LMP(fi::fp16, 1); break;
    case 2: if (v8) LMP(float, 4); else LMP(float, 1); break;
    default: return hipErrorInvalidValue;
  }
#undef LMP
  return hipGetLastError();
}

extern "C" hipError_t fi_mhc_pre(int dtype, int prenorm, const float* dot_mix,
                                 const float* sqrsum, const void* residual,
                                 const float* scale, const float* base,
                                 float* post_mix, float* comb_mix, void* layer_input,
                                 int64_t tokens, int H, int num_splits, float inv_k,
                                 float rms_eps, float pre_eps, float sink_eps,
                                 float post_mult, int sink_repeat,
                                 hipStream_t stream) {
  int grid = tokens < 4096 ? (int)tokens : 4096;
  if (grid == 0) return hipSuccess;
  dim3 g(grid), blk(256);
#define LMQ(T, V, PN)                                                             \
  hipLaunchKernelGGL((fi::mhc_pre_kernel<T, V, PN>), g, blk, 0, stream, dot_mix,  \
                     sqrsum, (const T*)residual, scale, base, post_mix, comb_mix, \
                     (T*)layer_input, tokens, H, num_splits, inv_k, rms_eps,      \
                     pre_eps, sink_eps, post_mult, sink_repeat)
#define LMQ2(T, V)                            \
  do {                                        \
    if (prenorm) LMQ(T, V, true);             \
    else LMQ(T, V, false);                    \
  } while (0)
  bool v8 = (H % 8 == 0);
  switch (dtype) {
    case 0: if (v8) LMQ2(fi::bf16, 8); else LMQ2(fi::bf16, 1); break;
    case 1: if (v8) LMQ2(fi::fp16, 8); else LMQ2(fi::fp16, 1); break;
    case 2: if (v8) LMQ2(float, 4); else LMQ2(float, 1); break;
    default: return hipErrorInvalidValue;
  }
#undef LMQ2
#undef LMQ
  return hipGetLastError();
}

extern "C" hipError_t fi_concat_mla_k(int dtype, void* k, const void* k_nope,
                                      const void* k_rope, int64_t tokens, int Hk,
                                      int nope, int rope, hipStream_t stream) {
  int grid = tokens < 4096 ? (int)tokens : 4096;
  if (grid == 0) return hipSuccess;
  dim3 g(grid), blk(256);
#define LCK(T, V)                                                               \
  hipLaunchKernelGGL((fi::concat_mla_k_kernel<T, V>), g, blk, 0, stream, (T*)k, \
                     (const T*)k_nope, (const T*)k_rope, tokens, Hk, nope, rope)
  bool v8 = (nope % 8 == 0) && (rope % 8 == 0);
  switch (dtype) {
    case 0: if (v8) LCK(fi::bf16, 8); else LCK(fi::bf16, 1); break;
    case 1: if (v8) LCK(fi::fp16, 8); else LCK(fi::fp16, 1); break;
    case 3: if (v8) LCK(fi::fp8_e4m3, 8); else LCK(fi::fp8_e4m3, 1); break;
    default: return hipErrorInvalidValue;
  }
#undef LCK
  return hipGetLastError();
}
