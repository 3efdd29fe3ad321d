// Mamba-2 selective state update (decode step) for gfx950. Parity with
// reference include/flashinfer/mamba/selective_state_update.cuh +
// csrc/selective_state_update.cu:
//   dt' = softplus(dt + dt_bias)            (optional)
//   state[b,h,p,s] = state * exp(dt' * A[h]) + dt' * x[b,h,p] * B[b,g,s]
//   y[b,h,p] = sum_s state[b,h,p,s] * C[b,g,s]  (+ D[h] * x)  (* silu(z))
// One thread per (b, h, p) row; the dstate axis is a sequential vectorized
// loop (state rows are contiguous). ngroups maps kv-style B/C sharing.
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/vec.hpp"

namespace fi {


template <typename T, bool STATE_F32>
__global__ void ssu_kernel(SSUParams p) {
  int64_t total = (int64_t)p.batch * p.nheads * p.headdim;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int pp = (int)(idx % p.headdim);
    int64_t bh = idx / p.headdim;
    int h = (int)(bh % p.nheads);
    int64_t b = bh / p.nheads;
    int g = h / (p.nheads / p.ngroups);

    float dt = to_f32<T>(((const T*)p.dt)[b * p.nheads + h]);
    if (p.dt_bias) dt += to_f32<T>(((const T*)p.dt_bias)[h]);
    if (p.dt_softplus) dt = dt > 20.f ? dt : __builtin_logf(1.f + __builtin_expf(dt));
    float a = to_f32<T>(((const T*)p.A)[h]);
    float dA = __builtin_expf(dt * a);
    float xv = to_f32<T>(((const T*)p.x)[(b * p.nheads + h) * p.headdim + pp]);
    float dBx = dt * xv;

    const T* Brow = (const T*)p.Bm + (b * p.ngroups + g) * p.dstate;
    const T* Crow = (const T*)p.Cm + (b * p.ngroups + g) * p.dstate;
    float y = 0.f;
    int64_t srow = ((b * p.nheads + h) * (int64_t)p.headdim + pp) * p.dstate;
    if constexpr (STATE_F32) {
      float* st = (float*)p.state + srow;
      for (int s = 0; s < p.dstate; ++s) {
        float ns = st[s] * dA + dBx * to_f32<T>(Brow[s]);
        st[s] = ns;
        y += ns * to_f32<T>(Crow[s]);
      }
    } else {
      T* st = (T*)p.state + srow;
      for (int s = 0; s < p.dstate; ++s) {
        float ns = to_f32<T>(st[s]) * dA + dBx * to_f32<T>(Brow[s]);
        st[s] = from_f32<T>(ns);
        y += ns * to_f32<T>(Crow[s]);
      }
    }
    if (p.D) y += to_f32<T>(((const T*)p.D)[h]) * xv;
    if (p.z) {
      float zv = to_f32<T>(((const T*)p.z)[(b * p.nheads + h) * p.headdim + pp]);
      y *= zv / (1.f + __builtin_expf(-zv));
    }
    ((T*)p.out)[(b * p.nheads + h) * p.headdim + pp] = from_f32<T>(y);
  }
}

// ---------------------------------------------------------------------------
// Mamba-2 SSD chunked prefill (reference parity: flashinfer/mamba/
// ssd_combined.py SSDCombined:250 / mamba_chunk_scan_combined semantics).
// Same recurrence as the decode step, scanned over a whole sequence:
//   dt'_t = clamp(softplus(dt_t + dt_bias), dt_limit)
//   S     = exp(dt'_t * A[h]) * S + dt'_t * x_t (x) B_t
//   y_t   = S C_t + D * x_t          (* silu(z_t))
// One block per (batch, head), blockDim = headdim; thread p owns the full
// [dstate] state row in registers, so the update and the y reduction are both
// thread-local. B/C rows and the per-token dt scalars stage through LDS in
// chunks of CH tokens.
template <typename T, int DSTATE>
__global__ __launch_bounds__(256, 2) void ssd_scan_kernel(
    const T* __restrict__ x, const float* __restrict__ dt,
    const float* __restrict__ A, const T* __restrict__ Bm, const T* __restrict__ Cm,
    const float* __restrict__ D, const T* __restrict__ z,
    const float* __restrict__ dt_bias, const float* __restrict__ init_states,
    float* __restrict__ final_states, T* __restrict__ out, int L, int H, int G,
    int P, int dt_softplus, float dt_min, float dt_max, int d_has_hdim) {
  constexpr int CH = 16;
  __shared__ float B_s[CH][DSTATE], C_s[CH][DSTATE];
  __shared__ float dA_s[CH], dtv_s[CH];
  const int b = blockIdx.x / H, h = blockIdx.x % H;
  const int g = h / (H / G);
  const int p = threadIdx.x;
  const bool live = p < P;  // blockDim is padded to a wave multiple; padded
                            // threads still help stage LDS and hit barriers

  float S[DSTATE];
  const int64_t st_off = (((int64_t)b * H + h) * P + (live ? p : 0)) * DSTATE;
#pragma unroll
  for (int s = 0; s < DSTATE; ++s)
    S[s] = (init_states && live) ? init_states[st_off + s] : 0.f;
  const float a = A[h];
  const float dval = (D && live) ? D[d_has_hdim ? h * P + p : h] : 0.f;

  for (int base = 0; base < L; base += CH) {
    const int nt = min(CH, L - base);
    __syncthreads();
    for (int e = threadIdx.x; e < nt * DSTATE; e += blockDim.x) {
      int t = e / DSTATE, s = e % DSTATE;
      int64_t row = (((int64_t)b * L + base + t) * G + g) * DSTATE + s;
      B_s[t][s] = to_f32<T>(Bm[row]);
      C_s[t][s] = to_f32<T>(Cm[row]);
    }
    for (int t = threadIdx.x; t < nt; t += blockDim.x) {
      float d = dt[((int64_t)b * L + base + t) * H + h];
      if (dt_bias) d += dt_bias[h];
      if (dt_softplus) d = d > 20.f ? d : __builtin_logf(1.f + __builtin_expf(d));
      d = fminf(fmaxf(d, dt_min), dt_max);
      dtv_s[t] = d;
      dA_s[t] = __builtin_expf(d * a);
    }
    __syncthreads();
    for (int t = 0; live && t < nt; ++t) {
      const int64_t xrow = (((int64_t)b * L + base + t) * H + h) * P + p;
      const float xv = to_f32<T>(x[xrow]);
      const float dBx = dtv_s[t] * xv, dA = dA_s[t];
      float y = 0.f;
#pragma unroll
      for (int s = 0; s < DSTATE; ++s) {
        S[s] = __builtin_fmaf(dBx, B_s[t][s], S[s] * dA);
        y = __builtin_fmaf(S[s], C_s[t][s], y);
      }
      y += dval * xv;
      if (z) {
        float zv = to_f32<T>(z[xrow]);
        y *= zv / (1.f + __builtin_expf(-zv));
      }
      out[xrow] = from_f32<T>(y);
    }
  }
  if (final_states && live)
#pragma unroll
    for (int s = 0; s < DSTATE; ++s) final_states[st_off + s] = S[s];
}

}  // namespace fi

extern "C" hipError_t fi_ssd_scan(int dtype, const void* x, const float* dt,
                                  const float* A, const void* Bm, const void* Cm,
                                  const float* D, const void* z, const float* dt_bias,
                                  const float* init_states, float* final_states,
                                  void* out, int batch, int L, int H, int G, int P,
                                  int dstate, int dt_softplus, float dt_min,
                                  float dt_max, int d_has_hdim, hipStream_t stream) {
  dim3 grid((uint32_t)batch * H), blk(((P + 63) / 64) * 64);
  if ((int)blk.x > 256) return hipErrorInvalidValue;
#define LSS(T, N)                                                                  \
  hipLaunchKernelGGL((fi::ssd_scan_kernel<T, N>), grid, blk, 0, stream,            \
                     (const T*)x, dt, A, (const T*)Bm, (const T*)Cm, D,            \
                     (const T*)z, dt_bias, init_states, final_states, (T*)out, L,  \
                     H, G, P, dt_softplus, dt_min, dt_max, d_has_hdim)
#define LSS2(T)                                    \
  do {                                             \
    if (dstate == 128) LSS(T, 128);                \
    else if (dstate == 64) LSS(T, 64);             \
    else return hipErrorInvalidValue;              \
  } while (0)
  switch (dtype) {
    case 0: LSS2(fi::bf16); break;
    case 1: LSS2(fi::fp16); break;
    case 2: LSS2(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LSS2
#undef LSS
  return hipGetLastError();
}

extern "C" hipError_t fi_selective_state_update(int dtype, fi::SSUParams* p,
                                                hipStream_t stream) {
  int64_t total = (int64_t)p->batch * p->nheads * p->headdim;
  int grid = (int)((total + 255) / 256);
  if (grid > 4096) grid = 4096;
  if (grid == 0) grid = 1;
  dim3 g(grid), blk(256);
#define LS(T, SF) hipLaunchKernelGGL((fi::ssu_kernel<T, SF>), g, blk, 0, stream, *p)
  switch (dtype * 2 + (p->state_f32 ? 1 : 0)) {
    case 0: LS(fi::bf16, false); break;
    case 1: LS(fi::bf16, true); break;
    case 2: LS(fi::fp16, false); break;
    case 3: LS(fi::fp16, true); break;
    case 4: LS(float, false); break;
    case 5: LS(float, true); break;
    default: return hipErrorInvalidValue;
  }
#undef LS
  return hipGetLastError();
}
