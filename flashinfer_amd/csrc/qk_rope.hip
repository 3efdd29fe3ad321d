// Fused per-head RMSNorm + interleaved RoPE over packed QKV (reference
// parity: flashinfer/norm/__init__.py fused_qk_rmsnorm_rope:1608; math per
// the reference's tests/norm/test_fused_qk_rmsnorm_rope.py). One wave per
// (token, head): 2 elems/lane, wave-local RMS reduction, cos/sin from the
// host-precomputed [S, D] tables (3D frame/height/width factorization done
// once in Python — the kernel is pure elementwise + reduce).
//   q/k heads: y = rope(rmsnorm(x) * w);   v heads: copy.
#include "fi/common.hpp"
#include "fi/vec.hpp"

namespace fi {

// interleaved rope on an even/odd pair (d, d+1):
//   y[d]   = x[d] * cos[d]   - x[d+1] * sin[d+1]
//   y[d+1] = x[d] * sin[d+1] + x[d+1] * cos[d]
template <typename T>
__global__ __launch_bounds__(256, 4) void qk_rope_kernel(
    const T* __restrict__ qkv,   // [B*S, (Hq+Hk+Hv) * D]
    const T* __restrict__ qw,    // [D]
    const T* __restrict__ kw,    // [D]
    const float* __restrict__ cos_t,  // [S, D]
    const float* __restrict__ sin_t,  // [S, D]
    T* __restrict__ q_out, T* __restrict__ k_out, T* __restrict__ v_out,
    int64_t tokens, int S, int Hq, int Hk, int Hv, int D, float eps,
    float attn_factor, int qk_norm) {
  const int H = Hq + Hk + Hv;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  for (int64_t tok = blockIdx.x; tok < tokens; tok += gridDim.x) {
    const int s = (int)(tok % S);
    const T* row = qkv + tok * (int64_t)H * D;
    for (int h = wid; h < H; h += 4) {       // one wave per head
      const bool is_q = h < Hq;
      const bool is_v = h >= Hq + Hk;
      const T* src = row + (int64_t)h * D;
      // exactly 2 elems per lane (D == 128 enforced by the launcher: the
      // wave RMS reduction needs every lane populated)
      {
        int d0 = lane * 2;
        float x0 = to_f32<T>(src[d0]);
        float x1 = to_f32<T>(src[d0 + 1]);
        float y0 = x0, y1 = x1;
        if (!is_v) {
          if (qk_norm) {
            float ss = x0 * x0 + x1 * x1;
            // wave sum over the head's D elems
            ss = wave_reduce_sum<kWaveSize>(ss);
            float rr = rsqrtf(ss / D + eps);
            const T* w = is_q ? qw : kw;
            y0 = x0 * rr * to_f32<T>(w[d0]);
            y1 = x1 * rr * to_f32<T>(w[d0 + 1]);
          }
          float c = cos_t[(int64_t)s * D + d0];
          float sn = sin_t[(int64_t)s * D + d0 + 1];
          float r0 = y0 * c - y1 * sn;
          float r1 = y0 * sn + y1 * c;
          y0 = r0 * attn_factor;
          y1 = r1 * attn_factor;
        }
        T* dst;
        if (is_q) dst = q_out + (tok * (int64_t)Hq + h) * D;
        else if (is_v) dst = v_out + (tok * (int64_t)Hv + (h - Hq - Hk)) * D;
        else dst = k_out + (tok * (int64_t)Hk + (h - Hq)) * D;
        dst[d0] = from_f32<T>(y0);
        dst[d0 + 1] = from_f32<T>(y1);
      }
    }
  }
}

}  // namespace fi

extern "C" hipError_t fi_qk_rope(int dtype, const void* qkv, const void* qw,
                                 const void* kw, const float* cos_t,
                                 const float* sin_t, void* q_out, void* k_out,
                                 void* v_out, int64_t tokens, int S, int Hq, int Hk,
                                 int Hv, int D, float eps, float attn_factor,
                                 int qk_norm, hipStream_t stream) {
  if (D != 128) return hipErrorInvalidValue;
  int grid = tokens < 8192 ? (int)tokens : 8192;
  if (grid == 0) return hipSuccess;
  dim3 g((uint32_t)grid), blk(256);
#define LQR(T)                                                                   \
  hipLaunchKernelGGL((fi::qk_rope_kernel<T>), g, blk, 0, stream, (const T*)qkv,  \
                     (const T*)qw, (const T*)kw, cos_t, sin_t, (T*)q_out,        \
                     (T*)k_out, (T*)v_out, tokens, S, Hq, Hk, Hv, D, eps,        \
                     attn_factor, qk_norm)
  switch (dtype) {
    case 0: LQR(fi::bf16); break;
    case 1: LQR(fi::fp16); break;
    default: return hipErrorInvalidValue;
  }
#undef LQR
  return hipGetLastError();
}
