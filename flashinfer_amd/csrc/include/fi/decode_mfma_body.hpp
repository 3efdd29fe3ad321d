// MFMA decode item body for gfx950 — shared by decode_mfma.hip and the
// persistent holistic batch_attention.hip (design narrative + reference
// citations in those TUs).
#pragma once
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/fastdiv.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"
#include "fi/prefill_body.hpp"  // kLog2e, b16x4, ds_read_tr16 helpers
#include "fi/vec.hpp"

namespace fi {

constexpr int kDecKVB = 32;       // kv tokens per 32x32 S^T MFMA pass
constexpr int kDecWaves = 8;      // 2 waves/SIMD at occupancy 1
constexpr int kDecVTileStride = 72;

// LDS arena bytes (per-wave K/V stage regions + Qs + merge m/d rows)
template <typename T, int HEAD_DIM, int GROUP>
constexpr int decode_mfma_smem_bytes() {
  constexpr int QROWS = GROUP < 32 ? GROUP : 32;
  return kDecWaves * kDecKVB * HEAD_DIM * (int)sizeof(T) +
         kDecWaves * (kDecKVB / 4) * (HEAD_DIM / 16) * kDecVTileStride * (int)sizeof(T) +
         32 * HEAD_DIM * (int)sizeof(T) + 2 * kDecWaves * QROWS * (int)sizeof(float);
}

// One (req, kv_head[, z-chunk]) MFMA decode item. Shared by the standalone
// decode_mfma kernel and the persistent holistic BatchAttention kernel.
// TKV != T (fp8 e4m3 KV cache): tokens are dequantized to T during the LDS
// staging write (reference prefill.cuh:1150 repack design) — HBM KV bytes
// halve, the MFMA pipeline stays bf16/f16; k_scale folds into sm_scale and
// v_scale is applied host-side (same contract as the vector kernel).
template <typename T, int HEAD_DIM, int GROUP, typename TKV = T>
__device__ __forceinline__ void decode_mfma_item_body(const DecodeParams& p, int req,
                                                      int kv_head, int zidx,
                                                      char* smem,
                                                      int qo_row = -1) {
  // req indexes the page table; qo_row the q/o/lse rows (they differ in the
  // holistic ragged mixed batch where qo_row = qo_indptr[req])
  if (qo_row < 0) qo_row = req;
  constexpr int WAVES = kDecWaves;
  constexpr int KVB = kDecKVB;
  constexpr int KCH = HEAD_DIM / 16;   // k-chunks in QK^T
  constexpr int DT = HEAD_DIM / 32;    // d-tiles in PV / output
  constexpr int KROWB = HEAD_DIM * 2;  // K LDS row bytes
  constexpr int QROWS = GROUP < 32 ? GROUP : 32;
  constexpr int VTILE_STRIDE = kDecVTileStride;

  // per-wave LDS regions (single-buffered; in-wave LDS ordering makes the
  // read-then-overwrite safe), carved from the caller's arena. The
  // cross-wave merge state is aliased into each wave's OWN stage region
  // after its loop ends (the one barrier separates the phases) — 8 waves of
  // K+V stage already fill the 160 KB budget at HEAD_DIM 128.
  constexpr int kKEl = KVB * HEAD_DIM;
  constexpr int kVEl = (KVB / 4) * (HEAD_DIM / 16) * VTILE_STRIDE;
  char* const smem_ = (char*)__builtin_assume_aligned(smem, 128);
  T* const KsB = reinterpret_cast<T*>(smem_);
  T* const VsB = KsB + WAVES * kKEl;
  T* const Qs = VsB + WAVES * kVEl;
  float* const lds_mB = reinterpret_cast<float*>(Qs + 32 * HEAD_DIM);
  float* const lds_dB = lds_mB + WAVES * QROWS;
  auto Ks = [&](int w) { return KsB + w * kKEl; };
  auto Vs = [&](int w) { return VsB + w * kVEl; };
  auto lds_m = [&](int w, int g) -> float& { return lds_mB[w * QROWS + g]; };
  auto lds_d = [&](int w, int g) -> float& { return lds_dB[w * QROWS + g]; };
  // merge row (g) of wave w: first 16 q rows live in Ks[w], the rest in Vs[w]
  auto merge_row = [&](int w, int g) -> float* {
    return g < 16 ? reinterpret_cast<float*>(Ks(w)) + g * HEAD_DIM
                  : reinterpret_cast<float*>(Vs(w)) + (g - 16) * HEAD_DIM;
  };
  static_assert(QROWS <= 16 || (KVB / 4) * (HEAD_DIM / 16) * VTILE_STRIDE * 2 >=
                                   (QROWS - 16) * HEAD_DIM * 4,
                "merge rows exceed the aliased stage LDS");

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int lq = lane & 31;      // q column this lane owns in S^T
  const int khalf = (lane >> 5) * 8;


  int np = p.kv_indptr[req + 1] - p.kv_indptr[req];
  int64_t kv_len =
      np == 0 ? 0 : (int64_t)(np - 1) * p.page_size.d + p.kv_last_page_len[req];
  int64_t kv_lo = 0;
  if (p.window_left >= 0) {
    int64_t w = kv_len - 1 - p.window_left;
    if (w > 0) kv_lo = w;
  }
  // cross-WG split (gridDim.z): z-chunk of the KV first, then the
  // contiguous per-wave slice within it — both 32-token aligned
  const int split = p.split > 1 ? p.split : 1;
  int64_t total = kv_len - kv_lo;
  int64_t z_lo = kv_lo, z_hi = kv_len;
  if (split > 1) {
    int64_t per_z = ((total + (int64_t)split * KVB - 1) / ((int64_t)split * KVB)) * KVB;
    z_lo = kv_lo + (int64_t)zidx * per_z;
    z_hi = z_lo + per_z;
    if (z_hi > kv_len) z_hi = kv_len;
    if (z_lo > kv_len) z_lo = kv_len;
    total = z_hi - z_lo;
  }
  int64_t per_wave = ((total + WAVES * KVB - 1) / (WAVES * KVB)) * KVB;
  int64_t ws_ = z_lo + (int64_t)wave * per_wave;
  int64_t we_ = ws_ + per_wave;
  if (we_ > z_hi) we_ = z_hi;

  const int32_t* page_ids = p.kv_indices + p.kv_indptr[req];
  const TKV* kbase = (const TKV*)p.k_data;
  const TKV* vbase = (const TKV*)p.v_data;

  // ---- Q staged ONCE into LDS (B-operand fragments re-read per tile via
  // ds_read: lgkm-tracked, so they never wait on the in-flight HBM staging
  // loads the way a per-tile GLOBAL re-read would — vmcnt is an in-order
  // counter — and they cost no resident VGPRs the way qf registers would
  // (measured: global re-read 33 us, register qf 108-132 B/lane spill) ----
  using frag = typename mfma_ab_frag<T>::type;
  {
    constexpr int QUNITS = 32 * HEAD_DIM / 8;
    for (int u = tid; u < QUNITS; u += WAVES * 64) {
      int row = u / (HEAD_DIM / 8);
      int chunk8 = u % (HEAD_DIM / 8);
      vec_t<T, 8> qv;
      if (row < QROWS) {
        qv.load((const T*)p.q + (int64_t)qo_row * p.q_stride_n +
                (int64_t)(kv_head * GROUP + row) * p.q_stride_h + chunk8 * 8);
      } else {
        qv.fill(0.f);
      }
      *reinterpret_cast<shortx8*>(reinterpret_cast<char*>(Qs) +
                                  (KROWB == 256 ? swz256(row * KROWB + chunk8 * 16)
                                                : swz128(row * KROWB + chunk8 * 16))) =
          *reinterpret_cast<const shortx8*>(qv.data);
    }
    __syncthreads();
  }

  float m_run = -INFINITY, d_run = 0.f;
  floatx16 acc_o[DT];
#pragma unroll
  for (int i = 0; i < DT; ++i) acc_o[i] = {};

  const float scale2 = p.sm_scale * kLog2e;
  const float cap = p.logits_soft_cap;
  // ALiBi: slope per q head (this lane's column); bias = -slope*(pos - kv)
  float slope = 0.f;
  if (p.alibi)
    slope = __builtin_exp2f(-8.f * (kv_head * GROUP + lq + 1) / p.num_qo_heads);

  // ---- register-staged K/V tile loads (one tile in regs while the staged
  // tile computes; 2 waves/SIMD co-residency fills the vmcnt gaps) ----
  constexpr int S_ITER = KVB * HEAD_DIM / 8 / 64;
  vec_t<TKV, 8> kregA[S_ITER], vregA[S_ITER];
  auto stage_load = [&](int64_t kv0) {
#pragma unroll
    for (int it = 0; it < S_ITER; ++it) {
      int u = lane + it * 64;
      int row = u / (HEAD_DIM / 8);
      int chunk8 = u % (HEAD_DIM / 8);
      int64_t kvpos = kv0 + row;
      kregA[it].fill(0.f);
      vregA[it].fill(0.f);
      if (kvpos < we_) {
        uint32_t pg, entry;
        p.page_size.divmod((uint32_t)kvpos, pg, entry);
        int64_t off = (int64_t)page_ids[pg] * p.stride_page +
                      (int64_t)kv_head * p.stride_h + (int64_t)entry * p.stride_n +
                      chunk8 * 8;
        kregA[it].load(kbase + off);
        vregA[it].load(vbase + off);
      }
    }
  };
  auto stage_write = [&]() {
#pragma unroll
    for (int it = 0; it < S_ITER; ++it) {
      int u = lane + it * 64;
      int row = u / (HEAD_DIM / 8);
      int chunk8 = u % (HEAD_DIM / 8);
      shortx8 kw, vw;
      if constexpr (__is_same(T, TKV)) {
        kw = *reinterpret_cast<const shortx8*>(kregA[it].data);
        vw = *reinterpret_cast<const shortx8*>(vregA[it].data);
      } else {
        // fp8 -> T dequant on the staging write
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          reinterpret_cast<T*>(&kw)[j] = from_f32<T>(kregA[it].get(j));
          reinterpret_cast<T*>(&vw)[j] = from_f32<T>(vregA[it].get(j));
        }
      }
      *reinterpret_cast<shortx8*>(reinterpret_cast<char*>(Ks(wave)) +
                                  (KROWB == 256 ? swz256(row * KROWB + chunk8 * 16)
                                                : swz128(row * KROWB + chunk8 * 16))) = kw;
      *reinterpret_cast<shortx8*>(
          reinterpret_cast<char*>(Vs(wave)) +
          ((row >> 2) * (HEAD_DIM / 16) + (chunk8 >> 1)) * (VTILE_STRIDE * 2) +
          (row & 3) * 32 + (chunk8 & 1) * 16) = vw;
    }
  };

  auto process_tile = [&](int64_t kv0) {
    // ---- S^T = K * Q^T (both operand frags from LDS) ----
    floatx16 acc_s = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < KCH; ++c) {
      auto swz = [&](uint32_t x) { return KROWB == 256 ? swz256(x) : swz128(x); };
      frag qf = *reinterpret_cast<const frag*>(
          reinterpret_cast<const char*>(Qs) +
          swz(lq * KROWB + (c * 16 + khalf) * 2));
      frag kfrag = *reinterpret_cast<const frag*>(
          reinterpret_cast<const char*>(Ks(wave)) +
          swz(lq * KROWB + (c * 16 + khalf) * 2));
      acc_s = mfma_ab_frag<T>::mma32(kfrag, qf, acc_s);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + base-2 logits (decode: all q rows at position kv_len-1,
    // so the bounds are the same for every row; soft-cap/ALiBi are cheap
    // per-element ops on the 16 in-lane values) ----
    float pr[16];
    if (kv0 + KVB <= we_ && cap <= 0.f && !p.alibi) {
#pragma unroll
      for (int r = 0; r < 16; ++r) pr[r] = acc_s[r] * scale2;
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int64_t kv = kv0 + mfma32_cd_row(r, lane);
        float sv = acc_s[r] * p.sm_scale;
        if (cap > 0.f) sv = cap * tanhf(sv / cap);
        if (p.alibi) sv -= slope * (float)(kv_len - 1 - kv);
        sv *= kLog2e;
        pr[r] = (kv < we_) ? sv : -INFINITY;
      }
    }

    // ---- online softmax (defer-max; exchange with lane^32) ----
    float tmax = pr[0];
#pragma unroll
    for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, pr[r]);
    tmax = xhalf_max(tmax);
    bool defer = m_run != -INFINITY && __all(tmax - m_run <= 8.f);
    float m_new = defer ? m_run : fmaxf(m_run, tmax);
    float f, psum = 0.f;
    if (m_new == -INFINITY) {
#pragma unroll
      for (int r = 0; r < 16; ++r) pr[r] = 0.f;
      f = 1.f;
    } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        pr[r] = (pr[r] == -INFINITY) ? 0.f : __builtin_exp2f(pr[r] - m_new);
        psum += pr[r];
      }
      f = defer ? 1.f : __builtin_exp2f(m_run - m_new);
    }
    d_run = d_run * f + psum;
    if (!defer) {
      m_run = m_new;
#pragma unroll
      for (int i = 0; i < DT; ++i) {
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_o[i][r] *= f;
      }
    }

    // ---- P^T fragments: pack + lazy half-exchange (prefill idiom) ----
    uint32_t W[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // dtype-correct packed convert (f16 kernels pack f16 fragments —
      // was unconditionally bf16, a latent fp16 numerics bug)
      if constexpr (__is_same(T, fp16)) {
        asm("v_cvt_pkrtz_f16_f32 %0, %1, %2"
            : "=v"(W[j])
            : "v"(pr[2 * j]), "v"(pr[2 * j + 1]));
      } else {
        asm("v_cvt_pk_bf16_f32 %0, %1, %2"
            : "=v"(W[j])
            : "v"(pr[2 * j]), "v"(pr[2 * j + 1]));
      }
    }
    // half-exchange via v_permlane32_swap (guide T12): 4 VALU ops replace
    // 8 ds_bpermute + 8 selects (b0[0]=concat(W0.lo,W2.lo),
    // b0[2]=concat(W0.hi,W2.hi), ...)
    uint32_t b0[4], b1[4];
    permlane32_pair(W[0], W[2], b0[0], b0[2]);
    permlane32_pair(W[1], W[3], b0[1], b0[3]);
    permlane32_pair(W[4], W[6], b1[0], b1[2]);
    permlane32_pair(W[5], W[7], b1[1], b1[3]);

    // ---- O^T += V^T * P^T via HW transpose reads ----
    {
      const uint32_t vb = (uint32_t)(uintptr_t)Vs(wave) + (uint32_t)(lane & 15) * 8;
      const uint32_t tdsel = ((lane >> 4) & 1);
      const int kvb = khalf;  // 0 or 8: kv sub-rows of this half
#pragma unroll
      for (int i = 0; i < DT; ++i) {
        uint32_t a00 = vb + (((kvb >> 2)) * (HEAD_DIM / 16) + i * 2 + tdsel) *
                               (VTILE_STRIDE * 2);
        b16x4 r00 = ds_read_tr16(a00);
        b16x4 r01 = ds_read_tr16(a00 + (HEAD_DIM / 16) * (VTILE_STRIDE * 2));
        uint32_t a10 = a00 + 4 * (HEAD_DIM / 16) * (VTILE_STRIDE * 2);
        b16x4 r10 = ds_read_tr16(a10);
        b16x4 r11 = ds_read_tr16(a10 + (HEAD_DIM / 16) * (VTILE_STRIDE * 2));
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
        union {
          b16x4 h[2];
          frag f_;
        } u0, u1;
        u0.h[0] = r00;
        u0.h[1] = r01;
        u1.h[0] = r10;
        u1.h[1] = r11;
        acc_o[i] = mfma_ab_frag<T>::mma32(u0.f_, *reinterpret_cast<frag*>(b0),
                                          acc_o[i]);
        acc_o[i] = mfma_ab_frag<T>::mma32(u1.f_, *reinterpret_cast<frag*>(b1),
                                          acc_o[i]);
      }
    }
  };

  // ---- main loop: wait stage(n) -> issue loads(n+1) -> compute(n) ->
  // ds_write(n+1) (in-wave LDS ordering makes the single buffer safe) ----
  if (ws_ < we_) {
    stage_load(ws_);
    stage_write();
  }
  for (int64_t kv0 = ws_; kv0 < we_; kv0 += KVB) {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const bool have_next = kv0 + KVB < we_;
    if (have_next) stage_load(kv0 + KVB);
    process_tile(kv0);
    if (have_next) stage_write();
  }

  // ---- per-wave state -> merge rows (aliased into this wave's stage LDS;
  // every wave has finished reading its stage before writing here, and the
  // barrier below orders the cross-wave reads) ----
  float d_full = xhalf_sum(d_run);
  if (lq < QROWS) {
    float* orow = merge_row(wave, lq);
#pragma unroll
    for (int i = 0; i < DT; ++i) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int d = i * 32 + mfma32_cd_row(r, lane);
        orow[d] = acc_o[i][r];
      }
    }
    if ((lane >> 5) == 0) {
      lds_m(wave, lq) = m_run;
      lds_d(wave, lq) = d_full;
    }
  }
  __syncthreads();

  // ---- cross-wave merge; split==1 stores directly, else the per-z
  // NORMALIZED partial + base-2 lse goes to the merge workspace, merged
  // either by the merge_states launch or (when the host passes counters)
  // by the last-arriving z-WG below. (A DEVICE-scope fence variant
  // measured 4x slower than a second launch — profiles/README r02; the
  // counter variant needs no fence because the host only enables it when
  // all z-WGs of a (req, kv_head) share one XCD.) ----
  T* obase = (T*)p.o + (int64_t)qo_row * p.o_stride_n;
  for (int idx = tid; idx < QROWS * HEAD_DIM; idx += WAVES * 64) {
    int g = idx / HEAD_DIM;
    int d = idx % HEAD_DIM;
    float m_star = -INFINITY;
#pragma unroll
    for (int w = 0; w < WAVES; ++w) m_star = fmaxf(m_star, lds_m(w, g));
    float d_sum = 0.f, o_sum = 0.f;
    if (m_star != -INFINITY) {
#pragma unroll
      for (int w = 0; w < WAVES; ++w) {
        float sw = __builtin_exp2f(lds_m(w, g) - m_star);
        d_sum += lds_d(w, g) * sw;
        o_sum += merge_row(w, g)[d] * sw;
      }
    }
    int qh = kv_head * GROUP + g;
    float inv_d = d_sum > 0.f ? 1.f / d_sum : 0.f;
    if (split == 1) {
      obase[(int64_t)qh * p.o_stride_h + d] = from_f32<T>(o_sum * inv_d);
      if (d == 0 && p.lse)
        p.lse[(int64_t)qo_row * p.num_qo_heads + qh] =
            d_sum > 0.f ? m_star + __builtin_log2f(d_sum) : -INFINITY;
    } else {
      // merge_states layout: item = req*split + z over ALL Hq heads
      int64_t item = (int64_t)qo_row * split + zidx;
      p.tmp_v[(item * p.num_qo_heads + qh) * HEAD_DIM + d] = o_sum * inv_d;
      if (d == 0)
        p.tmp_s[item * p.num_qo_heads + qh] =
            d_sum > 0.f ? m_star + __builtin_log2f(d_sum) : -INFINITY;
    }
  }

  // ---- same-XCD in-kernel split merge. Precondition (host-gated):
  // (batch * num_kv_heads) % 8 == 0, so the dispatcher's strict
  // round-robin (measured: linear_workgroup_id % 8 == XCC_ID exactly —
  // scripts/probe/xcd_probe.hip) places every z-WG of one (req, kv_head)
  // on the SAME XCD. Partials are then coherent in that XCD's L2 and the
  // last-arriving WG merges them with plain loads + an agent-scope
  // counter — no device-scope fence (the fenced variant measured 4x
  // slower). The winner resets its counter, so plan-time zeroing
  // suffices and the path is hipGraph-safe. ----
  if (split > 1 && p.counters) {
    __shared__ int s_lastwg;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // partials in L2
    __syncthreads();
    if (tid == 0) {
      uint32_t* ctr = p.counters + (int64_t)req * p.num_kv_heads + kv_head;
      // RELAXED on purpose: acq_rel at agent scope emits buffer_wbl2 +
      // buffer_inv (cross-XCD L2 maintenance — the 4x-slower mechanism).
      // Within one XCD the vmcnt(0) drain above IS the release (stores are
      // at the shared L2 before the atomic issues, and the atomic lands at
      // that same L2 in order); the winner's merge loads hit the same L2.
      uint32_t prev = __hip_atomic_fetch_add(ctr, 1u, __ATOMIC_RELAXED,
                                             __HIP_MEMORY_SCOPE_AGENT);
      s_lastwg = (prev == (uint32_t)(split - 1));
      if (s_lastwg)
        __hip_atomic_store(ctr, 0u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
      asm volatile("" ::: "memory");  // compile-time order: loads after win
    }
    __syncthreads();
    if (!s_lastwg) return;
    const int64_t base_item = (int64_t)qo_row * split;
    // per-g weights computed once (not per output element): one thread per
    // g reads the split lse values, exp2-weights them into LDS, writes lse
    __shared__ float s_mw[32 * 8];  // [g][z] normalized weight
    if (tid < QROWS) {
      const int g = tid;
      const int qh = kv_head * GROUP + g;
      float m_star = -INFINITY, sz[8];
      for (int z = 0; z < split; ++z) {
        sz[z] = p.tmp_s[(base_item + z) * p.num_qo_heads + qh];
        m_star = fmaxf(m_star, sz[z]);
      }
      float wsum = 0.f;
      for (int z = 0; z < split; ++z) {
        float wz = (m_star == -INFINITY || sz[z] == -INFINITY)
                       ? 0.f
                       : __builtin_exp2f(sz[z] - m_star);
        s_mw[g * 8 + z] = wz;
        wsum += wz;
      }
      float inv = wsum > 0.f ? 1.f / wsum : 0.f;
      for (int z = 0; z < split; ++z) s_mw[g * 8 + z] *= inv;
      if (p.lse)
        p.lse[(int64_t)qo_row * p.num_qo_heads + qh] =
            wsum > 0.f ? m_star + __builtin_log2f(wsum) : -INFINITY;
    }
    __syncthreads();
    for (int idx = tid; idx < QROWS * HEAD_DIM; idx += WAVES * 64) {
      int g = idx / HEAD_DIM;
      int d = idx % HEAD_DIM;
      int qh = kv_head * GROUP + g;
      float osum = 0.f;
      for (int z = 0; z < split; ++z)
        osum += p.tmp_v[((base_item + z) * p.num_qo_heads + qh) * HEAD_DIM +
                        d] *
                s_mw[g * 8 + z];
      obase[(int64_t)qh * p.o_stride_h + d] = from_f32<T>(osum);
    }
  }
}


}  // namespace fi
