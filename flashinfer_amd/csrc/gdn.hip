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

// ---------------------------------------------------------------------------
// Chunked GDN/KDA prefill (reference parity: flashinfer/gdn_kernels/blackwell/
// gdn_prefill.py chunk_gated_delta_rule_sm100:137 semantics, validated against
// tests/gdn/reference_delta_rule.py blockwise_delta_rule:856). Per-token
// recurrence over each ragged sequence:
//   S   <- a_t * S                      (gate decay; scalar or per-channel)
//   u_t <- beta_t * (v_t - k_t^T S)
//   S   <- S + k_t (x) u_t
//   o_t <- scale * q_t^T S
// One block per (seq, head, column-slice); the [D, D/SPLIT] f32 state slice
// lives in registers — every step of the recurrence is column-local (the
// k^T S and q^T S reductions run over ROWS), so Dv splits across SPLIT
// independent blocks for small batch*heads. Thread t owns column
// (slice + t % CB), rows [(t/CB)*RPT, +RPT). Tokens stage through LDS in
// chunks of CH; the cross-row reductions combine GPC partials through LDS.
template <typename T, int D, bool PER_CHANNEL_GATE, int SPLIT = 1>
__global__ __launch_bounds__(256, 2) void gdn_chunk_kernel(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ gate, const float* __restrict__ beta,
    T* __restrict__ out, const int32_t* __restrict__ cu_seqlens,
    const float* __restrict__ init_state, float* __restrict__ final_state,
    float scale, int H) {
  constexpr int NT = 256;
  constexpr int CB = D / SPLIT;     // columns per block
  constexpr int GPC = NT / CB;      // thread groups per column
  constexpr int RPT = D / GPC;      // state rows per thread
  constexpr int CH = 16;            // tokens staged per LDS chunk
  __shared__ float kq_s[CH][2 * D];                       // k then q
  __shared__ float v_s[CH][CB];
  __shared__ float a_s[CH][PER_CHANNEL_GATE ? D : 1];
  __shared__ float b_s[CH];
  __shared__ float part[2][GPC][CB];                      // kS / o partials
  __shared__ float delta_s[CB];

  const int seq = blockIdx.x / (H * SPLIT);
  const int h = (blockIdx.x / SPLIT) % H;
  const int c0 = (blockIdx.x % SPLIT) * CB;               // column slice base
  const int64_t s0 = cu_seqlens[seq], s1 = cu_seqlens[seq + 1];
  const int len = (int)(s1 - s0);
  const int tid = threadIdx.x;
  const int col = c0 + tid % CB, grp = tid / CB;
  const int r0 = grp * RPT;

  float S[RPT];
  const int64_t state_off = ((int64_t)seq * H + h) * D * D;
#pragma unroll
  for (int i = 0; i < RPT; ++i)
    S[i] = init_state ? init_state[state_off + (int64_t)(r0 + i) * D + col] : 0.f;
  if (len == 0) {  // zero-length sequence: state passes through untouched
    if (final_state)
#pragma unroll
      for (int i = 0; i < RPT; ++i)
        final_state[state_off + (int64_t)(r0 + i) * D + col] = S[i];
    return;
  }

  for (int base = 0; base < len; base += CH) {
    const int nt = min(CH, len - base);
    // stage the chunk: k/q/v rows for this head, gate/beta scalars
    for (int e = tid; e < nt * D; e += NT) {
      int t = e / D, d = e % D;
      int64_t row = (s0 + base + t) * (int64_t)H + h;
      kq_s[t][d] = to_f32<T>(k[row * D + d]);
      kq_s[t][D + d] = scale * to_f32<T>(q[row * D + d]);
      if (d >= c0 && d < c0 + CB) v_s[t][d - c0] = to_f32<T>(v[row * D + d]);
      if constexpr (PER_CHANNEL_GATE) a_s[t][d] = gate[row * D + d];
    }
    for (int t = tid; t < nt; t += NT) {
      int64_t row = (s0 + base + t) * (int64_t)H + h;
      if constexpr (!PER_CHANNEL_GATE) a_s[t][0] = gate[row];
      b_s[t] = beta[row];
    }
    __syncthreads();

    for (int t = 0; t < nt; ++t) {
      // decay + k^T S partial over this thread's rows
      float acc = 0.f;
#pragma unroll
      for (int i = 0; i < RPT; ++i) {
        float a = PER_CHANNEL_GATE ? a_s[t][r0 + i] : a_s[t][0];
        S[i] *= a;
        acc = __builtin_fmaf(kq_s[t][r0 + i], S[i], acc);
      }
      part[0][grp][col - c0] = acc;
      __syncthreads();
      if (tid < CB) {
        float ks = 0.f;
#pragma unroll
        for (int g2 = 0; g2 < GPC; ++g2) ks += part[0][g2][tid];
        delta_s[tid] = b_s[t] * (v_s[t][tid] - ks);
      }
      __syncthreads();
      // rank-1 update + q^T S partial
      const float dlt = delta_s[col - c0];
      float oacc = 0.f;
#pragma unroll
      for (int i = 0; i < RPT; ++i) {
        S[i] = __builtin_fmaf(kq_s[t][r0 + i], dlt, S[i]);
        oacc = __builtin_fmaf(kq_s[t][D + r0 + i], S[i], oacc);
      }
      part[1][grp][col - c0] = oacc;
      __syncthreads();
      if (tid < CB) {
        float o = 0.f;
#pragma unroll
        for (int g2 = 0; g2 < GPC; ++g2) o += part[1][g2][tid];
        out[((s0 + base + t) * (int64_t)H + h) * D + c0 + tid] = from_f32<T>(o);
      }
      __syncthreads();
    }
  }
  if (final_state)
#pragma unroll
    for (int i = 0; i < RPT; ++i)
      final_state[state_off + (int64_t)(r0 + i) * D + col] = S[i];
}

}  // namespace fi

extern "C" hipError_t fi_gdn_chunk(int dtype, int per_channel_gate, const void* q,
                                   const void* k, const void* v, const float* gate,
                                   const float* beta, void* out,
                                   const int32_t* cu_seqlens, const float* init_state,
                                   float* final_state, float scale, int num_seqs,
                                   int H, int D, hipStream_t stream) {
  // split the Dv columns across blocks until the 256-CU chip has >= 2
  // blocks/CU to schedule (each block is sequential over the sequence)
  int64_t bh = (int64_t)num_seqs * H;
  int split = 1;
  while (split < 4 && bh * split * 2 < 512 && D / (split * 2) >= 32) split *= 2;
  dim3 grid((uint32_t)(bh * split)), blk(256);
#define LGC(T, D, PC, SP)                                                        \
  hipLaunchKernelGGL((fi::gdn_chunk_kernel<T, D, PC, SP>), grid, blk, 0, stream, \
                     (const T*)q, (const T*)k, (const T*)v, gate, beta, (T*)out, \
                     cu_seqlens, init_state, final_state, scale, H)
#define LGC_SP(T, D, PC)                                              \
  do {                                                                \
    if (split == 4) LGC(T, D, PC, 4);                                 \
    else if (split == 2) LGC(T, D, PC, 2);                            \
    else LGC(T, D, PC, 1);                                            \
  } while (0)
#define LGC2(T)                                                       \
  do {                                                                \
    if (D == 128 && per_channel_gate) LGC_SP(T, 128, true);           \
    else if (D == 128) LGC_SP(T, 128, false);                         \
    else if (D == 64 && per_channel_gate) LGC_SP(T, 64, true);        \
    else if (D == 64) LGC_SP(T, 64, false);                           \
    else return hipErrorInvalidValue;                                 \
  } while (0)
  switch (dtype) {
    case 0: LGC2(fi::bf16); break;
    case 1: LGC2(fi::fp16); break;
    case 2: LGC2(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LGC2
#undef LGC_SP
#undef LGC
  return hipGetLastError();
}

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
