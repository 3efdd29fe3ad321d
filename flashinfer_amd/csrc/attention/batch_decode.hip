// Batch decode attention over paged KV — the core serving decode path.
// Functional parity with reference include/flashinfer/attention/decode.cuh
// (BatchDecodeWithPagedKVCacheKernel:615) + scheduler split-KV, re-designed
// for CDNA4:
//   * 4-wave workgroups, one (work-item, kv-head) unit per wave (64-thread
//     workgroups hit the per-CU workgroup cap before the wave cap); within a
//     wave, lanes split into head_dim/VPL-lane columns x tokens. The bf16
//     GROUP<=4 path loads 32 B/lane (VPL=16) behind a 4-stage register ring
//     — PMC r01 showed it HBM-LATENCY-bound (MemUnitStalled ~0, VALUBusy
//     ~30%) at its 2-wave occupancy, so ~3 iterations stay in flight.
//   * QK dots use packed v_dot2_f32_bf16 (keeps K in bf16 registers, halves
//     the VALU op count of the convert+FMA form).
//   * GQA group (1/2/4/8 q heads per kv head) processed in-register against
//     one K read — the K/V bytes are the bound, q reuse is free.
//   * split-KV via host-planned work items (req, chunk); partials are written
//     f32 to workspace and always reduced by the merge kernel (count==1 is a
//     plain normalize+cast) — keeps the kernel hipGraph-capturable with a
//     fixed grid.
//   * online-softmax state merged across the wave by shfl_xor state exchange.
// Single-request decode reuses this kernel (batch=1, one/few work items).
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/page.hpp"
#include "fi/state.hpp"
#include "fi/vec.hpp"

namespace fi {


// VPL elems per lane along head_dim (8 = 16B, or 32B for the small-group
// wide path); LPT = head_dim/VPL lanes per token; TPW = 64/LPT tokens/wave.
// The wide path (VPL=16, HEAD_DIM>=128 && GROUP<=4) halves the per-token
// shfl/softmax/address overhead per loaded byte — decode is VALU-bound at
// the target batch sizes (PMC r01) — at the cost of ~2x the accumulator
// registers, so its occupancy target drops to 2-3 waves/SIMD.
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;
typedef __attribute__((ext_vector_type(2))) _Float16 f16x2_t;

template <typename T, int N>
__device__ __forceinline__ float qk_dot(const void* qv, const void* kv, float acc) {
  if constexpr (__is_same(T, bf16)) {
    const bf16x2_t* q2 = (const bf16x2_t*)qv;
    const bf16x2_t* k2 = (const bf16x2_t*)kv;
#pragma unroll
    for (int i = 0; i < N / 2; ++i)
      acc = __builtin_amdgcn_fdot2_f32_bf16(q2[i], k2[i], acc, false);
    return acc;
  } else if constexpr (__is_same(T, fp16)) {
    const f16x2_t* q2 = (const f16x2_t*)qv;
    const f16x2_t* k2 = (const f16x2_t*)kv;
#pragma unroll
    for (int i = 0; i < N / 2; ++i)
      acc = __builtin_amdgcn_fdot2(q2[i], k2[i], acc, false);
    return acc;
  } else {
    const T* q = (const T*)qv;
    const T* k = (const T*)kv;
#pragma unroll
    for (int i = 0; i < N; ++i) acc += to_f32<T>(q[i]) * to_f32<T>(k[i]);
    return acc;
  }
}

template <typename T, typename TKV, int HEAD_DIM, int GROUP>
struct decode_traits {
  // the wide path needs a packed dot2 pipeline (bf16 or f16); f32/fp8-KV
  // paths carry f32 temporaries that would spill at its register budget
  static constexpr bool wide = (__is_same(T, bf16) || __is_same(T, fp16)) &&
                               __is_same(T, TKV) && HEAD_DIM >= 128 && GROUP <= 4;
  static constexpr int vpl = wide ? 16 : 8;
  // narrow path: the GROUP-8/16 accumulators (o_acc[GROUP][8]) blow the
  // 128-VGPR budget of 4 waves/SIMD — drop the target instead of spilling
  static constexpr int occ = wide ? (GROUP >= 4 ? 2 : 3)
                                  : (GROUP >= 16 ? 2 : (GROUP >= 8 ? 3 : 4));
};

template <typename T, typename TKV, int HEAD_DIM, int GROUP, bool SOFT_CAP>
__global__ __launch_bounds__(
    256,
    (decode_traits<T, TKV, HEAD_DIM, GROUP>::occ)) void batch_decode_kernel(DecodeParams p) {
  constexpr bool kSameT = __is_same(T, TKV);
  constexpr int VPL = decode_traits<T, TKV, HEAD_DIM, GROUP>::vpl;
  constexpr int LPT = HEAD_DIM / VPL;        // lanes per token
  constexpr int TPW = kWaveSize / LPT;       // tokens per wave
  const int lane = threadIdx.x & 63;
  const int tsub = lane / LPT;               // which token this lane covers
  const int dcol = (lane % LPT) * VPL;       // feature offset

  int64_t unit = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  int item = (int)(unit / p.num_kv_heads);
  int kv_head = (int)(unit % p.num_kv_heads);
  if (item >= p.n_items) return;
  int req = p.work_req[item];
  int chunk = p.work_chunk[item];

  int np = p.kv_indptr[req + 1] - p.kv_indptr[req];
  int64_t kv_len = np == 0 ? 0 : (int64_t)(np - 1) * p.page_size.d + p.kv_last_page_len[req];
  int64_t start = (int64_t)chunk * p.chunk_size;
  int64_t end = start + p.chunk_size;
  if (end > kv_len) end = kv_len;
  if (p.window_left >= 0) {
    int64_t w_start = kv_len - 1 - p.window_left;
    if (w_start > start) start = w_start;
  }

  // stage q packed (dot2 consumes bf16 pairs; sm_scale applied to the
  // reduced logit instead — 1 op per head per step). With a quantized KV
  // cache (TKV != T, e.g. fp8 e4m3) q is kept as f32 and the dot runs on
  // converted values.
  // GROUP>=8 narrow path: keeping GROUP q fragments resident is what pushes
  // the instantiation into loop-carried scratch spill (measured 104 B/lane,
  // dominating tiny-batch latency) — re-read q from L1 per step instead.
  constexpr bool kQReg = GROUP < 8 || decode_traits<T, TKV, HEAD_DIM, GROUP>::wide;
  constexpr int QR = kQReg ? GROUP : 1;
  vec_t<T, VPL> qreg[QR];
  float qf32[kSameT ? 1 : QR][kSameT ? 1 : VPL];
  const T* qbase = (const T*)p.q + (int64_t)req * p.q_stride_n +
                   (int64_t)(kv_head * GROUP) * p.q_stride_h + dcol;
  const float scale = p.sm_scale;
  if constexpr (kQReg) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      qreg[g].load(qbase + (int64_t)g * p.q_stride_h);
      if constexpr (!kSameT) {
#pragma unroll
        for (int j = 0; j < VPL; ++j) qf32[g][j] = qreg[g].get(j);
      }
    }
  }

  // base-2-domain online softmax with the single-FMA update:
  //   s <= m:  p = exp2(s - m); d += p; o += p*v          (8 FMA)
  //   s >  m:  r = exp2(m - s); d = d*r + 1; o = fma(o, r, v); m = s
  float m_run[GROUP], d_run[GROUP], o_acc[GROUP][VPL];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m_run[g] = -INFINITY;
    d_run[g] = 0.f;
#pragma unroll
    for (int j = 0; j < VPL; ++j) o_acc[g][j] = 0.f;
  }
  const float scale2 = scale * 1.4426950408889634f;

  const int32_t* page_ids = p.kv_indices + p.kv_indptr[req];
  const TKV* kbase = (const TKV*)p.k_data;
  const TKV* vbase = (const TKV*)p.v_data;

  // software-pipelined K/V loads hide the ~1us HBM latency under the
  // dependent VALU chain (guide T14 async-split idea, register-staged).
  auto addr_of = [&](int64_t pos0) -> int64_t {
    int64_t pos = pos0 + tsub;
    int64_t ppos = pos < end ? pos : (end - 1);
    uint32_t page_iter, entry;
    p.page_size.divmod((uint32_t)ppos, page_iter, entry);
    return (int64_t)page_ids[page_iter] * p.stride_page +
           (int64_t)kv_head * p.stride_h + (int64_t)entry * p.stride_n + dcol;
  };
  // STAGES-deep register ring (static indices via the unrolled sub-loop):
  // at 2-3 waves/SIMD the wide path has to keep >= 3 iterations of K/V in
  // flight to cover the ~1us HBM latency (PMC r01: MemUnitStalled ~0,
  // VALUBusy ~30% at 2-deep — latency-, not bandwidth- or VALU-bound).
  constexpr int STAGES = VPL == 16 ? (GROUP >= 4 ? 4 : 3) : 2;
  vec_t<TKV, VPL> kvb[STAGES], vvb[STAGES];
  auto process = [&](const vec_t<TKV, VPL>& kv_cur, const vec_t<TKV, VPL>& vv_cur,
                     int64_t pos0) {
    bool valid = pos0 + tsub < end;
    float vf[VPL];
#pragma unroll
    for (int j = 0; j < VPL; ++j) vf[j] = vv_cur.get(j);

#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      float s;
      if constexpr (kSameT) {
        if constexpr (kQReg) {
          s = qk_dot<T, VPL>(qreg[g].data, kv_cur.data, 0.f);
        } else {
          vec_t<T, VPL> qv;  // L1-resident: same 16 B every iteration
          qv.load(qbase + (int64_t)g * p.q_stride_h);
          s = qk_dot<T, VPL>(qv.data, kv_cur.data, 0.f);
        }
      } else {
        if constexpr (kQReg) {
          s = 0.f;
#pragma unroll
          for (int j = 0; j < VPL; ++j) s += qf32[g][j] * kv_cur.get(j);
        } else {
          vec_t<T, VPL> qv;
          qv.load(qbase + (int64_t)g * p.q_stride_h);
          s = 0.f;
#pragma unroll
          for (int j = 0; j < VPL; ++j) s += qv.get(j) * kv_cur.get(j);
        }
      }
      // reduce across the LPT lanes of this token
#pragma unroll
      for (int off2 = LPT / 2; off2 > 0; off2 >>= 1) s += __shfl_xor(s, off2, 64);
      float s2;
      if constexpr (SOFT_CAP) {
        s2 = p.logits_soft_cap * tanhf(s * scale / p.logits_soft_cap) *
             1.4426950408889634f;
      } else {
        s2 = s * scale2;
      }
      if (p.alibi) {
        // recomputed on demand: keeping GROUP slopes resident costs the
        // registers that push the GROUP-8/16 instantiations into scratch
        float slope2 = __builtin_exp2f(-8.f * (kv_head * GROUP + g + 1) /
                                       p.num_qo_heads) *
                       1.4426950408889634f;
        s2 -= slope2 * (float)(kv_len - 1 - (pos0 + tsub));
      }
      if (valid) {
        if (s2 <= m_run[g]) {
          float pv = __builtin_exp2f(s2 - m_run[g]);
          d_run[g] += pv;
#pragma unroll
          for (int j = 0; j < VPL; ++j) o_acc[g][j] += pv * vf[j];
        } else {
          float r = __builtin_exp2f(m_run[g] - s2);  // exp2(-inf)=0 first time
          d_run[g] = d_run[g] * r + 1.f;
#pragma unroll
          for (int j = 0; j < VPL; ++j) o_acc[g][j] = __builtin_fmaf(o_acc[g][j], r, vf[j]);
          m_run[g] = s2;
        }
      }
    }
  };
  if constexpr (STAGES == 2) {
    // classic copy-forward double buffer (lowest register pressure)
    if (start < end) {
      int64_t off = addr_of(start);
      kvb[0].load(kbase + off);
      vvb[0].load(vbase + off);
    }
    for (int64_t pos0 = start; pos0 < end; pos0 += TPW) {
      if (pos0 + TPW < end) {
        int64_t off = addr_of(pos0 + TPW);
        kvb[1].load(kbase + off);
        vvb[1].load(vbase + off);
      }
      process(kvb[0], vvb[0], pos0);
      kvb[0] = kvb[1];
      vvb[0] = vvb[1];
    }
  } else {
#pragma unroll
    for (int s = 0; s < STAGES - 1; ++s) {
      if (start + s * TPW < end) {
        int64_t off = addr_of(start + s * TPW);
        kvb[s].load(kbase + off);
        vvb[s].load(vbase + off);
      }
    }
    for (int64_t pos0 = start; pos0 < end; pos0 += (int64_t)TPW * STAGES) {
#pragma unroll
      for (int s = 0; s < STAGES; ++s) {
        int64_t cur = pos0 + s * TPW;
        if (cur >= end) break;
        int64_t pf = cur + (int64_t)(STAGES - 1) * TPW;
        if (pf < end) {
          int64_t off = addr_of(pf);
          kvb[(s + STAGES - 1) % STAGES].load(kbase + off);
          vvb[(s + STAGES - 1) % STAGES].load(vbase + off);
        }
        process(kvb[s], vvb[s], cur);
      }
    }
  }

  // merge the TPW per-token states across the wave: lanes with equal (lane%LPT)
  // hold the same output slice; exchange via shfl_xor (base-2 merge)
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
#pragma unroll
    for (int w = LPT; w < kWaveSize; w <<= 1) {
      float m_o = __shfl_xor(m_run[g], w, 64);
      float d_o = __shfl_xor(d_run[g], w, 64);
      float o_o[VPL];
#pragma unroll
      for (int j = 0; j < VPL; ++j) o_o[j] = __shfl_xor(o_acc[g][j], w, 64);
      float m_new = fmaxf(m_run[g], m_o);
      if (m_new != -INFINITY) {
        float s1 = __builtin_exp2f(m_run[g] - m_new);
        float s2x = __builtin_exp2f(m_o - m_new);
        d_run[g] = d_run[g] * s1 + d_o * s2x;
#pragma unroll
        for (int j = 0; j < VPL; ++j) o_acc[g][j] = o_acc[g][j] * s1 + o_o[j] * s2x;
        m_run[g] = m_new;
      }
    }
  }

  // lanes of token-sub 0 write the result (normalized partial + base-2 lse)
  if (tsub == 0) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      int qh = kv_head * GROUP + g;
      float inv_d = d_run[g] > 0.f ? 1.f / d_run[g] : 0.f;
      float* vout = p.tmp_v + ((int64_t)item * p.num_qo_heads + qh) * HEAD_DIM + dcol;
#pragma unroll
      for (int j = 0; j < VPL; ++j) vout[j] = o_acc[g][j] * inv_d;
      if (lane == 0)
        p.tmp_s[(int64_t)item * p.num_qo_heads + qh] =
            d_run[g] > 0.f ? m_run[g] + __builtin_log2f(d_run[g]) : -INFINITY;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused whole-request decode: the short-kv / small-batch latency shape
// (reference decode.cuh:615 + work-estimation scheduler.cuh:150 role, but a
// different algorithm: no host split, no merge kernel, no f32 tmp round-trip).
//
// One workgroup per (request, kv_head); 8 waves each take a contiguous slice
// of the request's KV, run the same register-ring online-softmax pipeline,
// then the 8 partial states are merged through LDS and the final output is
// written directly in the q dtype — ONE kernel launch end-to-end. At the
// BASELINE bs=16/kv=1024 GQA-8 config the split path pays two launches, a
// 2048-item f32 partial round-trip and a 1-wave-per-item grid; here the same
// work is 128 workgroups x 8 resident waves with a 4-deep load ring.
//
// The inner loop is intentionally a sibling of batch_decode_kernel's, not a
// shared function: this shape runs at launch_bounds(512,1) (256-VGPR budget,
// GROUP-8 keeps q resident and a 4-deep ring with no spill) while the split
// kernel is tuned at (256, 2-4); folding them would regress the tuned path.
template <typename T, typename TKV, int HEAD_DIM, int GROUP, bool SOFT_CAP,
          bool WIDE8 = false>
__global__ __launch_bounds__(512, 1) void decode_fused_kernel(DecodeParams p) {
  constexpr bool kSameT = __is_same(T, TKV);
  // GROUP-8 wide (WIDE8): tried — VPL=16 would halve the per-token
  // shfl-reduce chain, but o_acc[8][16] + the ring spills 191 VGPRs even at
  // launch_bounds(512,1); never instantiated (GQA-8 small-batch decode is
  // the MFMA decode kernel's job instead).
  constexpr bool kWide = (__is_same(T, bf16) || __is_same(T, fp16)) && kSameT &&
                         HEAD_DIM >= 128 && (GROUP <= 4 || (WIDE8 && GROUP == 8));
  constexpr int VPL = kWide ? 16 : 8;
  constexpr int LPT = HEAD_DIM / VPL;
  constexpr int TPW = kWaveSize / LPT;
  constexpr int WAVES = 8;
  constexpr int STAGES = (kWide && GROUP >= 8) ? 2 : 4;
  constexpr bool kQReg = !(kWide && GROUP >= 8);
  __shared__ float lds_o[WAVES][GROUP][HEAD_DIM];
  __shared__ float lds_m[WAVES][GROUP];
  __shared__ float lds_d[WAVES][GROUP];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tsub = lane / LPT;
  const int dcol = (lane % LPT) * VPL;
  const int req = blockIdx.x;
  const int kv_head = blockIdx.y;

  int np = p.kv_indptr[req + 1] - p.kv_indptr[req];
  int64_t kv_len =
      np == 0 ? 0 : (int64_t)(np - 1) * p.page_size.d + p.kv_last_page_len[req];
  int64_t start = 0;
  if (p.window_left >= 0) {
    int64_t w_start = kv_len - 1 - p.window_left;
    if (w_start > 0) start = w_start;
  }
  // contiguous per-wave slice of [start, kv_len)
  int64_t total = kv_len - start;
  int64_t per_wave = (total + WAVES - 1) / WAVES;
  int64_t ws = start + (int64_t)wave * per_wave;
  int64_t end = ws + per_wave;
  if (end > kv_len) end = kv_len;

  constexpr int QR = kQReg ? GROUP : 1;
  vec_t<T, VPL> qreg[QR];
  float qf32[kSameT ? 1 : QR][kSameT ? 1 : VPL];
  const T* qbase = (const T*)p.q + (int64_t)req * p.q_stride_n +
                   (int64_t)(kv_head * GROUP) * p.q_stride_h + dcol;
  if constexpr (kQReg) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      qreg[g].load(qbase + (int64_t)g * p.q_stride_h);
      if constexpr (!kSameT) {
#pragma unroll
        for (int j = 0; j < VPL; ++j) qf32[g][j] = qreg[g].get(j);
      }
    }
  }
  const float scale = p.sm_scale;
  const float scale2 = scale * 1.4426950408889634f;

  float m_run[GROUP], d_run[GROUP], o_acc[GROUP][VPL];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m_run[g] = -INFINITY;
    d_run[g] = 0.f;
#pragma unroll
    for (int j = 0; j < VPL; ++j) o_acc[g][j] = 0.f;
  }

  const int32_t* page_ids = p.kv_indices + p.kv_indptr[req];
  const TKV* kbase = (const TKV*)p.k_data;
  const TKV* vbase = (const TKV*)p.v_data;
  auto addr_of = [&](int64_t pos0) -> int64_t {
    int64_t pos = pos0 + tsub;
    int64_t ppos = pos < end ? pos : (end - 1);
    uint32_t page_iter, entry;
    p.page_size.divmod((uint32_t)ppos, page_iter, entry);
    return (int64_t)page_ids[page_iter] * p.stride_page +
           (int64_t)kv_head * p.stride_h + (int64_t)entry * p.stride_n + dcol;
  };
  vec_t<TKV, VPL> kvb[STAGES], vvb[STAGES];
  auto process = [&](const vec_t<TKV, VPL>& kv_cur, const vec_t<TKV, VPL>& vv_cur,
                     int64_t pos0) {
    bool valid = pos0 + tsub < end;
    float vf[VPL];
#pragma unroll
    for (int j = 0; j < VPL; ++j) vf[j] = vv_cur.get(j);
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      float s;
      if constexpr (kSameT) {
        if constexpr (kQReg) {
          s = qk_dot<T, VPL>(qreg[g].data, kv_cur.data, 0.f);
        } else {
          vec_t<T, VPL> qv;  // L1-resident: same bytes every iteration
          qv.load(qbase + (int64_t)g * p.q_stride_h);
          s = qk_dot<T, VPL>(qv.data, kv_cur.data, 0.f);
        }
      } else {
        s = 0.f;
#pragma unroll
        for (int j = 0; j < VPL; ++j) s += qf32[g][j] * kv_cur.get(j);
      }
#pragma unroll
      for (int off2 = LPT / 2; off2 > 0; off2 >>= 1) s += __shfl_xor(s, off2, 64);
      float s2;
      if constexpr (SOFT_CAP) {
        s2 = p.logits_soft_cap * tanhf(s * scale / p.logits_soft_cap) *
             1.4426950408889634f;
      } else {
        s2 = s * scale2;
      }
      if (p.alibi) {
        float slope2 = __builtin_exp2f(-8.f * (kv_head * GROUP + g + 1) /
                                       p.num_qo_heads) *
                       1.4426950408889634f;
        s2 -= slope2 * (float)(kv_len - 1 - (pos0 + tsub));
      }
      if (valid) {
        if (s2 <= m_run[g]) {
          float pv = __builtin_exp2f(s2 - m_run[g]);
          d_run[g] += pv;
#pragma unroll
          for (int j = 0; j < VPL; ++j) o_acc[g][j] += pv * vf[j];
        } else {
          float r = __builtin_exp2f(m_run[g] - s2);
          d_run[g] = d_run[g] * r + 1.f;
#pragma unroll
          for (int j = 0; j < VPL; ++j)
            o_acc[g][j] = __builtin_fmaf(o_acc[g][j], r, vf[j]);
          m_run[g] = s2;
        }
      }
    }
  };
#pragma unroll
  for (int s = 0; s < STAGES - 1; ++s) {
    if (ws + s * TPW < end) {
      int64_t off = addr_of(ws + s * TPW);
      kvb[s].load(kbase + off);
      vvb[s].load(vbase + off);
    }
  }
  for (int64_t pos0 = ws; pos0 < end; pos0 += (int64_t)TPW * STAGES) {
#pragma unroll
    for (int s = 0; s < STAGES; ++s) {
      int64_t cur = pos0 + s * TPW;
      if (cur >= end) break;
      int64_t pf = cur + (int64_t)(STAGES - 1) * TPW;
      if (pf < end) {
        int64_t off = addr_of(pf);
        kvb[(s + STAGES - 1) % STAGES].load(kbase + off);
        vvb[(s + STAGES - 1) % STAGES].load(vbase + off);
      }
      process(kvb[s], vvb[s], cur);
    }
  }

  // wave-internal merge (after this every lane holds the wave-merged state)
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
#pragma unroll
    for (int w = LPT; w < kWaveSize; w <<= 1) {
      float m_o = __shfl_xor(m_run[g], w, 64);
      float d_o = __shfl_xor(d_run[g], w, 64);
      float o_o[VPL];
#pragma unroll
      for (int j = 0; j < VPL; ++j) o_o[j] = __shfl_xor(o_acc[g][j], w, 64);
      float m_new = fmaxf(m_run[g], m_o);
      if (m_new != -INFINITY) {
        float s1 = __builtin_exp2f(m_run[g] - m_new);
        float s2x = __builtin_exp2f(m_o - m_new);
        d_run[g] = d_run[g] * s1 + d_o * s2x;
#pragma unroll
        for (int j = 0; j < VPL; ++j) o_acc[g][j] = o_acc[g][j] * s1 + o_o[j] * s2x;
        m_run[g] = m_new;
      }
    }
  }
  // stage the 8 per-wave states through LDS
  if (tsub == 0) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
#pragma unroll
      for (int j = 0; j < VPL; ++j) lds_o[wave][g][dcol + j] = o_acc[g][j];
      if (lane == 0) {
        lds_m[wave][g] = m_run[g];
        lds_d[wave][g] = d_run[g];
      }
    }
  }
  __syncthreads();
  // cross-wave merge + direct store: thread t owns output element (g, d)
  T* obase = (T*)p.o + (int64_t)req * p.o_stride_n;
  for (int idx = threadIdx.x; idx < GROUP * HEAD_DIM; idx += WAVES * kWaveSize) {
    int g = idx / HEAD_DIM;
    int d = idx % HEAD_DIM;
    float m_star = -INFINITY;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) m_star = fmaxf(m_star, lds_m[w][g]);
    float d_sum = 0.f, o_sum = 0.f;
    if (m_star != -INFINITY) {
#pragma unroll
      for (int w = 0; w < WAVES; ++w) {
        float sw = __builtin_exp2f(lds_m[w][g] - m_star);
        d_sum += lds_d[w][g] * sw;
        o_sum += lds_o[w][g][d] * sw;
      }
    }
    int qh = kv_head * GROUP + g;
    float inv_d = d_sum > 0.f ? 1.f / d_sum : 0.f;
    obase[(int64_t)qh * p.o_stride_h + d] = from_f32<T>(o_sum * inv_d);
    if (d == 0 && p.lse)
      p.lse[(int64_t)req * p.num_qo_heads + qh] =
          d_sum > 0.f ? m_star + __builtin_log2f(d_sum) : -INFINITY;
  }
}

template <typename T, typename TKV>
hipError_t decode_fused_dispatch(DecodeParams& p, hipStream_t stream) {
  int group = p.num_qo_heads / p.num_kv_heads;
  dim3 g((uint32_t)p.batch, (uint32_t)p.num_kv_heads), blk(512);
  bool sc = p.logits_soft_cap > 0.f;
#define LAUNCH_F(HD, G, SC) \
  hipLaunchKernelGGL((decode_fused_kernel<T, TKV, HD, G, SC>), g, blk, 0, stream, p)
#define DISPATCH_FG(HD, SC)                             \
  do {                                                  \
    switch (group) {                                    \
      case 1: LAUNCH_F(HD, 1, SC); break;               \
      case 2: LAUNCH_F(HD, 2, SC); break;               \
      case 4: LAUNCH_F(HD, 4, SC); break;               \
      case 8: LAUNCH_F(HD, 8, SC); break;               \
      default: return hipErrorInvalidValue;             \
    }                                                   \
  } while (0)
  if (p.head_dim == 64) { if (sc) DISPATCH_FG(64, true); else DISPATCH_FG(64, false); }
  else if (p.head_dim == 128) { if (sc) DISPATCH_FG(128, true); else DISPATCH_FG(128, false); }
  else if (p.head_dim == 256 && group <= 4) {
    // GROUP-8 x HD-256 LDS footprint (65 KB) exceeds the 64 KB workgroup
    // budget — the planner routes that combination to the split path
    if (sc) { switch (group) { case 1: LAUNCH_F(256,1,true); break; case 2: LAUNCH_F(256,2,true); break; case 4: LAUNCH_F(256,4,true); break; } }
    else { switch (group) { case 1: LAUNCH_F(256,1,false); break; case 2: LAUNCH_F(256,2,false); break; case 4: LAUNCH_F(256,4,false); break; } }
  } else return hipErrorInvalidValue;
#undef DISPATCH_FG
#undef LAUNCH_F
  return hipGetLastError();
}

template <typename T, typename TKV>
hipError_t decode_dispatch(DecodeParams& p, hipStream_t stream) {
  int group = p.num_qo_heads / p.num_kv_heads;
  int64_t units = (int64_t)p.n_items * p.num_kv_heads;
  dim3 g((uint32_t)((units + 3) / 4)), blk(256);
  bool sc = p.logits_soft_cap > 0.f;
#define LAUNCH_D(HD, G, SC) \
  hipLaunchKernelGGL((batch_decode_kernel<T, TKV, HD, G, SC>), g, blk, 0, stream, p)
#define DISPATCH_G(HD, SC)                              \
  do {                                                  \
    switch (group) {                                    \
      case 1: LAUNCH_D(HD, 1, SC); break;               \
      case 2: LAUNCH_D(HD, 2, SC); break;               \
      case 4: LAUNCH_D(HD, 4, SC); break;               \
      case 5: LAUNCH_D(HD, 5, SC); break;               \
      case 6: LAUNCH_D(HD, 6, SC); break;               \
      case 7: LAUNCH_D(HD, 7, SC); break;               \
      case 8: LAUNCH_D(HD, 8, SC); break;               \
      case 16: LAUNCH_D(HD, 16, SC); break;             \
      default: return hipErrorInvalidValue;             \
    }                                                   \
  } while (0)
#define DISPATCH_HD(SC)                                 \
  do {                                                  \
    switch (p.head_dim) {                               \
      case 64: DISPATCH_G(64, SC); break;               \
      case 128: DISPATCH_G(128, SC); break;             \
      case 256: DISPATCH_G(256, SC); break;             \
      default: return hipErrorInvalidValue;             \
    }                                                   \
  } while (0)
  if (sc) DISPATCH_HD(true);
  else DISPATCH_HD(false);
#undef DISPATCH_HD
#undef DISPATCH_G
#undef LAUNCH_D
  return hipGetLastError();
}

}  // namespace fi

// dtype = q dtype (0 bf16, 1 f16, 2 f32); kv_dtype additionally 3 = fp8 e4m3
extern "C" hipError_t fi_batch_decode(int dtype, int kv_dtype, fi::DecodeParams* p,
                                      hipStream_t stream) {
  if (p->n_items == 0) return hipSuccess;
  if (kv_dtype == dtype) {
    switch (dtype) {
      case 0: return fi::decode_dispatch<fi::bf16, fi::bf16>(*p, stream);
      case 1: return fi::decode_dispatch<fi::fp16, fi::fp16>(*p, stream);
      case 2: return fi::decode_dispatch<float, float>(*p, stream);
    }
  } else if (kv_dtype == 3 && dtype == 0) {
    return fi::decode_dispatch<fi::bf16, fi::fp8_e4m3>(*p, stream);
  } else if (kv_dtype == 3 && dtype == 1) {
    return fi::decode_dispatch<fi::fp16, fi::fp8_e4m3>(*p, stream);
  }
  return hipErrorInvalidValue;
}

extern "C" hipError_t fi_batch_decode_fused(int dtype, int kv_dtype,
                                            fi::DecodeParams* p, hipStream_t stream) {
  if (p->batch == 0) return hipSuccess;
  if (kv_dtype == dtype) {
    switch (dtype) {
      case 0: return fi::decode_fused_dispatch<fi::bf16, fi::bf16>(*p, stream);
      case 1: return fi::decode_fused_dispatch<fi::fp16, fi::fp16>(*p, stream);
      case 2: return fi::decode_fused_dispatch<float, float>(*p, stream);
    }
  } else if (kv_dtype == 3 && dtype == 0) {
    return fi::decode_fused_dispatch<fi::bf16, fi::fp8_e4m3>(*p, stream);
  } else if (kv_dtype == 3 && dtype == 1) {
    return fi::decode_fused_dispatch<fi::fp16, fi::fp8_e4m3>(*p, stream);
  }
  return hipErrorInvalidValue;
}
