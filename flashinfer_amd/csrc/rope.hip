// Rotary positional embedding for gfx950. Parity with reference
// include/flashinfer/pos_enc.cuh (BatchQKApplyRotaryPosIdsKernel:650,
// cos-sin-cache variant:361, llama-3.1 scaling:1528). One launch rotates both
// q [nnz, Hq, D] and k [nnz, Hkv, D] given pos_ids [nnz].
//
// Two paths:
//  * cos_sin_cache: f32 cache [max_pos, rot_dim] = [cos(rot/2) | sin(rot/2)]
//    (host-precomputed — the fast path; on-device trig turns this
//    memory-bound op VALU-bound).
//  * on-the-fly: freqs from rope_theta with optional llama-3.1 wavelength
//    scaling, computed per (lane, dim-pair) once and reused across tokens.
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/vec.hpp"

namespace fi {


// PVEC consecutive rotation pairs per thread.
template <typename T, int PVEC, bool kUseCache, bool kInterleave>
__global__ void rope_kernel(RopeParams p) {
  int half = p.rot_dim / 2;
  int pairs_per_head = half / PVEC;
  int heads = p.num_qo_heads + p.num_kv_heads;
  int64_t total = p.nnz * heads * pairs_per_head;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int pair0 = (int)(idx % pairs_per_head) * PVEC;
    int64_t th = idx / pairs_per_head;
    int head = (int)(th % heads);
    int64_t token = th / heads;
    int pos = p.pos_ids[token];

    float cosv[PVEC], sinv[PVEC];
#pragma unroll
    for (int j = 0; j < PVEC; ++j) {
      int i = pair0 + j;  // rotation pair index in [0, half)
      if constexpr (kUseCache) {
        cosv[j] = p.cos_sin_cache[(int64_t)pos * p.rot_dim + i];
        sinv[j] = p.cos_sin_cache[(int64_t)pos * p.rot_dim + half + i];
      } else {
        float freq = __builtin_expf(-2.f * i / p.rot_dim * __builtin_logf(p.rope_theta));
        if (p.smooth_a != 0.f || p.smooth_b != 0.f) {
          // llama-3.1: smooth interp between freq/factor and freq
          float smooth = fmaxf(0.f, fminf(1.f, freq * p.smooth_a + p.smooth_b));
          freq = (1.f - smooth) * (freq * p.rcp_factor) + smooth * freq;
        } else {
          freq *= p.rope_rcp_scale;
        }
        float ang = pos * freq;
        __sincosf(ang, &sinv[j], &cosv[j]);
      }
    }

    bool is_q = head < p.num_qo_heads;
    const T* src;
    T* dst;
    if (is_q) {
      src = (const T*)p.q + token * p.q_stride_n + (int64_t)head * p.q_stride_h;
      dst = (T*)p.q_out + token * p.o_q_stride_n + (int64_t)head * p.o_q_stride_h;
    } else {
      int kh = head - p.num_qo_heads;
      src = (const T*)p.k + token * p.k_stride_n + (int64_t)kh * p.k_stride_h;
      dst = (T*)p.k_out + token * p.o_k_stride_n + (int64_t)kh * p.o_k_stride_h;
    }

    if constexpr (kInterleave) {
      // pairs are (2i, 2i+1)
#pragma unroll
      for (int j = 0; j < PVEC; ++j) {
        int i = pair0 + j;
        float x1 = to_f32<T>(src[2 * i]);
        float x2 = to_f32<T>(src[2 * i + 1]);
        dst[2 * i] = from_f32<T>(x1 * cosv[j] - x2 * sinv[j]);
        dst[2 * i + 1] = from_f32<T>(x2 * cosv[j] + x1 * sinv[j]);
      }
    } else {
      // pairs are (i, i + half): vectorized PVEC-wide loads from both halves
      vec_t<T, PVEC> v1, v2, o1, o2;
      v1.load(src + pair0);
      v2.load(src + pair0 + half);
#pragma unroll
      for (int j = 0; j < PVEC; ++j) {
        float x1 = v1.get(j), x2 = v2.get(j);
        o1.set(j, x1 * cosv[j] - x2 * sinv[j]);
        o2.set(j, x2 * cosv[j] + x1 * sinv[j]);
      }
      o1.store(dst + pair0);
      o2.store(dst + pair0 + half);
    }
    // pass-through of dims beyond rot_dim (partial rotary)
    if (p.rot_dim < p.head_dim && dst != src) {
      for (int i = p.rot_dim + (int)(idx % pairs_per_head); i < p.head_dim;
           i += pairs_per_head) {
        dst[i] = src[i];
      }
    }
  }
}

template <typename T>
hipError_t rope_launch(const RopeParams& p, hipStream_t stream) {
  int half = p.rot_dim / 2;
  int pvec = (!p.interleave && half % 4 == 0) ? 4 : 1;
  int heads = p.num_qo_heads + p.num_kv_heads;
  int64_t total = p.nnz * heads * (half / pvec);
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
  if (grid == 0) grid = 1;
  dim3 g(grid), blk(256);
  bool cache = p.cos_sin_cache != nullptr;
#define LAUNCH_R(PV, C, I) \
  hipLaunchKernelGGL((rope_kernel<T, PV, C, I>), g, blk, 0, stream, p)
  if (p.interleave) {
    if (cache) LAUNCH_R(1, true, true);
    else LAUNCH_R(1, false, true);
  } else if (pvec == 4) {
    if (cache) LAUNCH_R(4, true, false);
    else LAUNCH_R(4, false, false);
  } else {
    if (cache) LAUNCH_R(1, true, false);
    else LAUNCH_R(1, false, false);
  }
#undef LAUNCH_R
  return hipGetLastError();
}

}  // namespace fi

extern "C" hipError_t fi_rope(int dtype, fi::RopeParams* p, hipStream_t stream) {
  switch (dtype) {
    case 0: return fi::rope_launch<fi::bf16>(*p, stream);
    case 1: return fi::rope_launch<fi::fp16>(*p, stream);
    case 2: return fi::rope_launch<float>(*p, stream);
  }
  return hipErrorInvalidValue;
}
