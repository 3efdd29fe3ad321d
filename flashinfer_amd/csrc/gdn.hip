// Gated DeltaNet (GDN) fused decode step for gfx950. Parity with reference
// flashinfer/gdn_kernels gdn_fused_decode_step (__init__.py:121) semantics:
//   S      <- g * S                       (per-head scalar gate decay)
//   v'     <- beta * (v - k^T S)          (delta rule correction)
//   S      <- S + k (x) v'                (rank-1 update)
//   o      <- q^T S
// State S: [B, H, Dk, Dv] (f32 or bf16); q/k: [B, H, Dk]; v/o: [B, H, Dv];
// g/beta: [B, H] f32.
// One block per (b, h); thread j owns state column j (coalesced across the
// Dv-contiguous rows); k/q staged in LDS; two passes over the column (the
// delta needs the full k^T S reduction before the update).
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

// PER_CHANNEL_GATE: g is [B, H, Dk] (KDA-style diagonal decay) instead of a
// per-head scalar (GDN).
template <typename T, typename TS, bool PER_CHANNEL_GATE>
__global__ void gdn_decode_kernel(TS* __restrict__ state, const T* __restrict__ q,
                                  const T* __restrict__ k, const T* __restrict__ v,
                                  const float* __restrict__ g,
                                  const float* __restrict__ beta, T* __restrict__ out,
                                  int B, int H, int Dk, int Dv) {
  extern __shared__ float smem[];  // k [Dk], q [Dk], (per-channel) g [Dk]
  float* ks = smem;
  float* qs = smem + Dk;
  float* gs = smem + 2 * Dk;
  int64_t bh = blockIdx.x;
  if (bh >= (int64_t)B * H) return;
  const T* kr = k + bh * Dk;
  const T* qr = q + bh * Dk;
  for (int i = threadIdx.x; i < Dk; i += blockDim.x) {
    ks[i] = to_f32<T>(kr[i]);
    qs[i] = to_f32<T>(qr[i]);
    if constexpr (PER_CHANNEL_GATE) gs[i] = g[bh * Dk + i];
  }
  __syncthreads();
  float gv = PER_CHANNEL_GATE ? 0.f : g[bh];
  float bv = beta[bh];
  TS* S = state + bh * (int64_t)Dk * Dv;
  for (int j = threadIdx.x; j < Dv; j += blockDim.x) {
    // pass 1: delta_j = beta * (v_j - sum_i k_i * g_i * S_ij)
    float acc = 0.f;
    for (int i = 0; i < Dk; ++i) {
      float gi = PER_CHANNEL_GATE ? gs[i] : gv;
      acc += ks[i] * gi * to_f32<TS>(S[(int64_t)i * Dv + j]);
    }
    float delta = bv * (to_f32<T>(v[bh * Dv + j]) - acc);
    // pass 2: update + output
    float o = 0.f;
    for (int i = 0; i < Dk; ++i) {
      float gi = PER_CHANNEL_GATE ? gs[i] : gv;
      float s_new = gi * to_f32<TS>(S[(int64_t)i * Dv + j]) + ks[i] * delta;
      S[(int64_t)i * Dv + j] = from_f32<TS>(s_new);
      o += qs[i] * s_new;
    }
    out[bh * Dv + j] = from_f32<T>(o);
  }
}

}  // namespace fi

extern "C" hipError_t fi_gdn_decode(int dtype, int state_f32, int per_channel_gate,
                                    void* state, const void* q, const void* k,
                                    const void* v, const float* g, const float* beta,
                                    void* out, int B, int H, int Dk, int Dv,
                                    hipStream_t stream) {
  int threads = Dv < 256 ? ((Dv + 63) / 64) * 64 : 256;
  if (threads == 0) threads = 64;
  size_t smem = 3 * Dk * sizeof(float);
  dim3 grid((uint32_t)((int64_t)B * H)), blk(threads);
#define LGDN(T, TS, PC)                                                            \
  hipLaunchKernelGGL((fi::gdn_decode_kernel<T, TS, PC>), grid, blk, smem, stream,  \
                     (TS*)state, (const T*)q, (const T*)k, (const T*)v, g, beta,   \
                     (T*)out, B, H, Dk, Dv)
#define LGDN2(T, TS)                                                               \
  do {                                                                             \
    if (per_channel_gate) LGDN(T, TS, true);                                       \
    else LGDN(T, TS, false);                                                       \
  } while (0)
  switch (dtype * 2 + (state_f32 ? 1 : 0)) {
    case 0: LGDN2(fi::bf16, fi::bf16); break;
    case 1: LGDN2(fi::bf16, float); break;
    case 2: LGDN2(fi::fp16, fi::fp16); break;
    case 3: LGDN2(fi::fp16, float); break;
    case 4: LGDN2(float, float); break;
    case 5: LGDN2(float, float); break;
    default: return hipErrorInvalidValue;
  }
#undef LGDN2
#undef LGDN
  return hipGetLastError();
}
