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

}  // namespace fi

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
