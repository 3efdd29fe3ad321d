// FA2-class prefill tile body for gfx950 — shared by batch_prefill.hip and
// the persistent holistic batch_attention.hip (see those TUs for the design
// narrative and reference citations).
#pragma once
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/fastdiv.hpp"
#include "fi/frag.hpp"
#include "fi/mfma.hpp"
#include "fi/profiler.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int kVTileStride = 72;  // V subtile stride in elems (144 B)

// Attention-variant hook (reference variants.cuh:31 DefaultAttention /
// variant_helper.cuh:75 + the csrc/*_customize_config.jinja injection
// mechanism): user-supplied device code compiled into the kernel via
// flashinfer_amd.jit.gen_customize_batch_prefill_module. kActive=false
// keeps the default instantiations codegen-identical (the interior
// fast path stays).
struct NoVariant {
  static constexpr bool kActive = false;
  __device__ static float logits_transform(float s, int /*qo_idx*/,
                                           int64_t /*kv_idx*/, int /*head*/,
                                           int /*qo_len*/, int64_t /*kv_len*/) {
    return s;
  }
  __device__ static bool logits_mask(int /*qo_idx*/, int64_t /*kv_idx*/,
                                     int /*head*/, int /*qo_len*/,
                                     int64_t /*kv_len*/) {
    return true;
  }
};

constexpr int KVB = 64;      // prefill kv tile
constexpr float kLog2e = 1.4426950408889634f;

typedef __attribute__((ext_vector_type(4))) __bf16 b16x4;

// gfx950 LDS transpose read: each 16-lane group collectively reads a [4][16]
// 16-bit row-major subtile (lane addr = subtile_base + (lane&15)*8 bytes) and
// lane l receives column (l&15) — rows 0..3 in order.
__device__ __forceinline__ b16x4 ds_read_tr16(uint32_t lds_byte_off) {
  b16x4 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(r) : "v"(lds_byte_off));
  return r;
}

template <int ROWB>
__device__ __forceinline__ uint32_t swz_row(uint32_t byte_off) {
  // XOR swizzle for ROWB-byte rows
  if constexpr (ROWB == 256) return swz256(byte_off);
  else if constexpr (ROWB == 128) return swz128(byte_off);
  else return byte_off;  // 512B rows (D=256): every other row aliases; use 256-swizzle
}

// CTAQ = packed q rows per workgroup (128 -> 4 waves, 256 -> 8 waves; the
// planner picks by average packed length — bigger tiles amortize K/V staging
// over more q rows). HDQK/HDVO may differ (DeepSeek MHA 192/128, reference
// benchmarks/samples/sample_testlist_output.csv:5 config).
// LDS arena bytes this body needs (double-buffered K + V stages)
template <typename T, int HDQK, int HDVO>
constexpr int prefill_tile_smem_bytes() {
  constexpr int KROWB = HDQK == 192 ? 400 : HDQK * 2;
  return 2 * KVB * KROWB +
         2 * (KVB / 4) * (HDVO / 16) * kVTileStride * (int)sizeof(T);
}

// One (req, q_tile, kv_head[, kv_chunk]) prefill tile: the FA2 MFMA pipeline.
// chunk >= 0 selects split-KV partial mode (writes p.tmp_v/tmp_s slots).
// Shared by the standalone batch_prefill kernel and the persistent holistic
// BatchAttention kernel (reference attention/persistent.cuh role).
template <typename T, typename TKV, int HDQK, int HDVO, int CTAQ, bool PAGED,
          bool CAUSAL, bool MASK = false, typename VARIANT = NoVariant>
__device__ __forceinline__ void prefill_tile_body(const PrefillParams& p, int req,
                                                  int qstart, int kv_head, int chunk_in,
                                                  char* smem) {
  char* const smem_ = (char*)__builtin_assume_aligned(smem, 128);
  constexpr int NTHREADS = CTAQ * 2;
  // fp8 (e4m3) KV caches are dequantized to bf16 during the LDS staging
  // write (reference prefill.cuh:1150 repack_fp8_tile_to_bf16 design): the
  // HBM bytes halve, the MFMA pipeline stays bf16.
  constexpr bool kF8KV = !__is_same(T, TKV);
  constexpr int KCH = HDQK / 16;  // k-chunks in QK^T
  constexpr int DT = HDVO / 32;   // d-tiles in PV / output
  // K tile row stride: power-of-two dims keep the XOR swizzle; 192 (384 B
  // rows) pads to 400 B — row*400 mod 256 has period 16 rows, so 32-row
  // fragment reads see at worst 2-way bank conflicts with no XOR needed.
  constexpr int KROWB = HDQK == 192 ? 400 : HDQK * 2;

  // DOUBLE-BUFFERED K/V stage (carved from the caller's LDS arena): compute
  // reads buf while the next tile's registers write buf^1 — ONE barrier per
  // KV tile instead of two (PMC r01: barrier-, not MFMA-/bandwidth-bound).
  // V is stored in [KVB/4][HDVO/16] subtiles of [4 kv][16 d] (row-major,
  // 128 B each) padded to 144 B stride: the shape ds_read_b64_tr_b16 wants
  // (each 16-lane group reads one subtile and receives it transposed), with
  // the pad de-aliasing subtile bank positions.
  constexpr int VTILE_STRIDE = kVTileStride;  // elems (144 B)
  constexpr int kKElems = KVB * KROWB / 2;
  constexpr int kVElems = (KVB / 4) * (HDVO / 16) * VTILE_STRIDE;
  T* const KsB = reinterpret_cast<T*>(smem_);
  T* const VsB = KsB + 2 * kKElems;
  auto Ks = [&](int b) { return KsB + b * kKElems; };
  auto Vs = [&](int b) { return VsB + b * kVElems; };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int lq = lane & 31;      // this lane's q row within the wave tile
  const int khalf = (lane >> 5) * 8;


  const int qo_begin = p.qo_indptr[req];
  const int qo_len = p.qo_indptr[req + 1] - qo_begin;
  const uint32_t group = p.group.d;
  const int packed_len = qo_len * group;

  int64_t kv_len;
  const int32_t* page_ids = nullptr;
  int64_t kv_base = 0;
  if constexpr (PAGED) {
    int np = p.kv_indptr[req + 1] - p.kv_indptr[req];
    kv_len = np == 0 ? 0 : (int64_t)(np - 1) * p.page_size.d + p.kv_last_page_len[req];
    page_ids = p.kv_indices + p.kv_indptr[req];
  } else {
    kv_base = p.kv_indptr[req];
    kv_len = p.kv_indptr[req + 1] - kv_base;
  }

  // this wave's q rows: [qstart + wid*32, +32)
  const int wq0 = qstart + wid * 32;
  const int my_row = wq0 + lq;          // packed row of this lane
  uint32_t my_qpos_u, my_g_u;
  p.group.divmod((uint32_t)(my_row < packed_len ? my_row : 0), my_qpos_u, my_g_u);
  const int my_qpos = (int)my_qpos_u;
  const bool row_valid = my_row < packed_len;

  // causal offset: kv position of q row my_qpos's "diagonal"
  const int64_t diag = kv_len - qo_len;

  // split-KV: this tile covers only [chunk*kv_chunk, +kv_chunk) of the KV
  // (bounds-masking clamps to kv_valid_hi; causal/ALiBi geometry keeps the
  // true kv_len). n_chunks is derived per request from the plan's global
  // chunk size (reference scheduler.cuh:101 binary-searched kv_chunk_size).
  const int chunk = chunk_in < 0 ? 0 : chunk_in;
  int n_chunks = 1;
  int64_t kv_valid_hi = kv_len;
  if (chunk_in >= 0) {
    n_chunks = (int)((kv_len + p.kv_chunk - 1) / p.kv_chunk);
    if (n_chunks < 1) n_chunks = 1;
    int64_t chi = ((int64_t)chunk + 1) * p.kv_chunk;
    if (chi < kv_valid_hi) kv_valid_hi = chi;
  }

  // ---- load Q fragments, PRE-SCALED by sm_scale*log2e: S^T comes out of
  // the MFMA already in the base-2 logit domain, removing 16 VALU multiplies
  // per 32-kv subtile from the softmax chain (PMC r01: VALUBusy 42.7 vs
  // MfmaUtil 16.4 — the logits pipeline is the bound). bf16 precision is
  // relative, so pre-scaling costs no accuracy. The soft-cap/ALiBi/variant
  // branch un-scales where it needs the natural domain. ----
  using frag = typename mfma_ab_frag<T>::type;
  frag qf[KCH];
  {
    const T* qptr = (const T*)p.q +
                    (int64_t)(qo_begin + my_qpos) * p.q_stride_n +
                    (int64_t)(kv_head * group + my_g_u) * p.q_stride_h;
    const float qs = p.sm_scale * kLog2e;
#pragma unroll
    for (int c = 0; c < KCH; ++c) {
      if (row_valid) {
        qf[c] = *reinterpret_cast<const frag*>(qptr + c * 16 + khalf);
      } else {
        qf[c] = frag{};
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        reinterpret_cast<T*>(&qf[c])[j] =
            from_f32<T>(to_f32<T>(reinterpret_cast<T*>(&qf[c])[j]) * qs);
      }
    }
  }

  // online softmax state (per lane = per q row; m/d in base-2 domain)
  float m_run = -INFINITY;
  float d_run = 0.f;
  floatx16 acc_o[DT];
#pragma unroll
  for (int i = 0; i < DT; ++i) acc_o[i] = {};

  const float cap = p.logits_soft_cap;
  const int wleft = p.window_left;

  // kv range this tile must process (causal upper bound; window lower bound;
  // split-KV chunk bounds); each wave also skips compute past its own rows'
  // causal bound.
  int64_t kv_hi = kv_valid_hi;
  int64_t wave_kv_hi = kv_valid_hi;
  if constexpr (CAUSAL) {
    int tile_max_qpos = (qstart + CTAQ - 1) / (int)group;
    if (tile_max_qpos >= qo_len) tile_max_qpos = qo_len - 1;
    int64_t hi = tile_max_qpos + diag + 1;
    if (hi < kv_hi) kv_hi = hi;
    int wave_max_qpos = (wq0 + 31) / (int)group;
    if (wave_max_qpos >= qo_len) wave_max_qpos = qo_len - 1;
    wave_kv_hi = wave_max_qpos + diag + 1;
  }
  int64_t kv_lo = 0;
  if (chunk_in >= 0) kv_lo = (int64_t)chunk * p.kv_chunk;
  if (wleft >= 0) {
    int tile_min_qpos = qstart / (int)group;
    int64_t lo = tile_min_qpos + diag - wleft;
    if (lo > kv_lo) kv_lo = lo;
  }

  const TKV* kbase = (const TKV*)p.k_data;
  const TKV* vbase = (const TKV*)p.v_data;

  // ---- async-STAGE split (guide T14): global loads for tile n+1 are issued
  // BEFORE tile n's compute (HBM latency hides under the MFMA phases); the
  // LDS writes land between the two barriers after compute. ----
  constexpr int KS_ITER = KVB * HDQK / 8 / NTHREADS;
  constexpr int VS_ITER = KVB * HDVO / 8 / NTHREADS;
  vec_t<TKV, 8> kreg[KS_ITER], vreg[VS_ITER];

  auto stage_load = [&](int64_t kv0) {
    if constexpr (HDQK == HDVO) {
      // square dims: one page lookup serves both K and V (k/v caches share a
      // layout, so the k strides address both — the tuned flagship path)
#pragma unroll
      for (int it = 0; it < KS_ITER; ++it) {
        int u = tid + it * NTHREADS;
        int row = u / (HDQK / 8);
        int chunk8 = u % (HDQK / 8);
        int64_t kvpos = kv0 + row;
        kreg[it].fill(0.f);
        vreg[it].fill(0.f);
        if (kvpos < kv_valid_hi) {
          int64_t off;
          if constexpr (PAGED) {
            uint32_t pg, entry;
            p.page_size.divmod((uint32_t)kvpos, pg, entry);
            off = (int64_t)page_ids[pg] * p.kv_stride_page +
                  (int64_t)kv_head * p.kv_stride_h + (int64_t)entry * p.kv_stride_n +
                  chunk8 * 8;
          } else {
            off = (kv_base + kvpos) * p.kv_stride_n + (int64_t)kv_head * p.kv_stride_h +
                  chunk8 * 8;
          }
          kreg[it].load(kbase + off);
          vreg[it].load(vbase + off);
        }
      }
    } else {
#pragma unroll
      for (int it = 0; it < KS_ITER; ++it) {
        int u = tid + it * NTHREADS;
        int row = u / (HDQK / 8);
        int chunk8 = u % (HDQK / 8);
        int64_t kvpos = kv0 + row;
        kreg[it].fill(0.f);
        if (kvpos < kv_valid_hi) {
          int64_t off;
          if constexpr (PAGED) {
            uint32_t pg, entry;
            p.page_size.divmod((uint32_t)kvpos, pg, entry);
            off = (int64_t)page_ids[pg] * p.kv_stride_page +
                  (int64_t)kv_head * p.kv_stride_h + (int64_t)entry * p.kv_stride_n;
          } else {
            off = (kv_base + kvpos) * p.kv_stride_n + (int64_t)kv_head * p.kv_stride_h;
          }
          kreg[it].load(kbase + off + chunk8 * 8);
        }
      }
#pragma unroll
      for (int it = 0; it < VS_ITER; ++it) {
        int u = tid + it * NTHREADS;
        int row = u / (HDVO / 8);
        int chunk8 = u % (HDVO / 8);
        int64_t kvpos = kv0 + row;
        vreg[it].fill(0.f);
        if (kvpos < kv_valid_hi) {
          int64_t off;
          if constexpr (PAGED) {
            uint32_t pg, entry;
            p.page_size.divmod((uint32_t)kvpos, pg, entry);
            off = (int64_t)page_ids[pg] * p.v_stride_page +
                  (int64_t)kv_head * p.v_stride_h + (int64_t)entry * p.v_stride_n;
          } else {
            off = (kv_base + kvpos) * p.v_stride_n + (int64_t)kv_head * p.v_stride_h;
          }
          vreg[it].load(vbase + off + chunk8 * 8);
        }
      }
    }
  };
  auto stage_write = [&](int wb) {
#pragma unroll
    for (int it = 0; it < KS_ITER; ++it) {
      int u = tid + it * NTHREADS;
      int row = u / (HDQK / 8);
      int chunk = u % (HDQK / 8);
      shortx8 kw;
      if constexpr (kF8KV) {
        // dequantize fp8 -> bf16 on the staging write
#pragma unroll
        for (int j = 0; j < 8; ++j)
          reinterpret_cast<T*>(&kw)[j] = from_f32<T>(kreg[it].get(j) * p.k_descale);
      } else {
        kw = *reinterpret_cast<const shortx8*>(kreg[it].data);
      }
      // K: row-major swizzled (vector frag reads)
      *reinterpret_cast<shortx8*>(reinterpret_cast<char*>(Ks(wb)) +
                                  swz_row<KROWB>(row * KROWB + chunk * 16)) = kw;
    }
#pragma unroll
    for (int it = 0; it < VS_ITER; ++it) {
      int u = tid + it * NTHREADS;
      int row = u / (HDVO / 8);
      int chunk = u % (HDVO / 8);
      shortx8 vw;
      if constexpr (kF8KV) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          reinterpret_cast<T*>(&vw)[j] = from_f32<T>(vreg[it].get(j) * p.v_descale);
      } else {
        vw = *reinterpret_cast<const shortx8*>(vreg[it].data);
      }
      // V: tr-read subtiled
      *reinterpret_cast<shortx8*>(
          reinterpret_cast<char*>(Vs(wb)) +
          ((row >> 2) * (HDVO / 16) + (chunk >> 1)) * (VTILE_STRIDE * 2) +
          (row & 3) * 32 + (chunk & 1) * 16) = vw;
    }
  };

  prof_event(p.prof_buf, 0, ProfType::kBegin);  // event 0: whole tile
  stage_load(kv_lo);
  stage_write(0);
  __syncthreads();

  prof_event(p.prof_buf, 1, ProfType::kBegin);  // event 1: kv mainloop
  int buf = 0;
  for (int64_t kv0 = kv_lo; kv0 < kv_hi; kv0 += KVB) {
    const bool have_next = kv0 + KVB < kv_hi;
    if (have_next) stage_load(kv0 + KVB);

    if (kv0 < wave_kv_hi) {
#pragma unroll
    for (int kt = 0; kt < KVB / 32; ++kt) {
      // causal: skip a 32-kv chunk every one of this wave's rows masks out
      // (the diagonal KVB often has only its first half visible) —
      // wave_kv_hi is wave-uniform, so the branch is divergence-free
      if (CAUSAL && kv0 + kt * 32 >= wave_kv_hi) continue;
      // ---- S^T = K * Q^T : [32 kv][32 q] ----
      floatx16 acc_s = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int c = 0; c < KCH; ++c) {
        frag kfrag = *reinterpret_cast<const frag*>(
            reinterpret_cast<const char*>(Ks(buf)) +
            swz_row<KROWB>((kt * 32 + lq) * KROWB + (c * 16 + khalf) * 2));
        acc_s = mfma_ab_frag<T>::mma32(kfrag, qf[c], acc_s);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- masking + base-2 logits -> p values (in place) ----
      // interior fast path: when the whole 32-kv tile is in-bounds, inside
      // every causal/window bound for this wave's rows, and unmasked, the
      // per-element predicate work (the VALU hotspot) is skipped.
      float pr[16];
      const int64_t kvt0 = kv0 + kt * 32;
      bool tile_full = (kvt0 + 32 <= kv_valid_hi) && !MASK && cap <= 0.f &&
                       !p.alibi && !VARIANT::kActive;
      if constexpr (CAUSAL) {
        int wave_min_qpos = wq0 / (int)group;
        tile_full &= (kvt0 + 32 <= wave_min_qpos + diag + 1);
      }
      if (wleft >= 0) {
        int wave_max_qpos2 = (wq0 + 31) / (int)group;
        tile_full &= (kvt0 >= wave_max_qpos2 + diag - wleft);
      }
      if (tile_full) {
#pragma unroll
        for (int r = 0; r < 16; ++r) pr[r] = acc_s[r];  // already base-2 domain
      } else {
        // ALiBi slope for this lane's qo head: 2^(-8*(h+1)/Hq)
        float slope = 0.f;
        if (p.alibi) {
          int qh = (int)(kv_head * group + my_g_u);
          slope = __builtin_exp2f(-8.f * (qh + 1) / p.num_qo_heads);
        }
        constexpr float kInvLog2e = 0.6931471805599453f;  // 1/log2(e)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int64_t kv = kvt0 + mfma32_cd_row(r, lane);
          float sv = acc_s[r];  // base-2 domain (Q pre-scaled)
          if (cap > 0.f)
            sv = cap * tanhf(sv * kInvLog2e / cap) * kLog2e;
          if (p.alibi) sv -= (slope * kLog2e) * (float)(my_qpos + diag - kv);
          bool ok = kv < kv_valid_hi;
          if constexpr (VARIANT::kActive) {
            int head = (int)(kv_head * group + my_g_u);
            sv = VARIANT::logits_transform(sv * kInvLog2e, my_qpos, kv, head,
                                           qo_len, kv_len) * kLog2e;
            ok &= VARIANT::logits_mask(my_qpos, kv, head, qo_len, kv_len);
          }
          if constexpr (CAUSAL) ok &= kv <= my_qpos + diag;
          if (wleft >= 0) ok &= kv >= my_qpos + diag - wleft;
          if constexpr (MASK) {
            if (ok) {
              int64_t bit = (int64_t)my_qpos * kv_len + kv;
              uint8_t byte = p.mask_data[p.mask_byte_indptr[req] + (bit >> 3)];
              ok &= (byte >> (bit & 7)) & 1;
            }
          }
          pr[r] = ok ? sv : -INFINITY;
        }
      }

      // ---- online softmax update (per lane; exchange with lane^32).
      // packed-f32 max/sum trees (v_pk_max/add_f32) halve the reduction
      // VALU (PMC r02: VALUBusy 47% — the softmax chain IS the bound) ----
      typedef __attribute__((ext_vector_type(2))) float f32x2_;
      const f32x2_* pr2 = reinterpret_cast<const f32x2_*>(pr);
      f32x2_ mx2 = pr2[0];
#pragma unroll
      for (int r = 1; r < 8; ++r) mx2 = __builtin_elementwise_max(mx2, pr2[r]);
      float tmax = fmaxf(mx2.x, mx2.y);
      tmax = xhalf_max(tmax);
      // defer-max (guide T13): skip the O-wide rescale while the running max
      // grows by < 8 (base-2) — p values stay bounded by 2^8, f32 accum is
      // fine. __all keeps the wave branch-uniform.
      bool defer = m_run != -INFINITY && __all(tmax - m_run <= 8.f);
      float m_new = defer ? m_run : fmaxf(m_run, tmax);
      float f, psum = 0.f;
      if (m_new == -INFINITY) {
#pragma unroll
        for (int r = 0; r < 16; ++r) pr[r] = 0.f;
        f = 1.f;
      } else {
        // no -inf select: v_exp_f32(-inf - finite) = 0 in hardware, and the
        // m_new == -inf case is handled above (16 cndmask saved per subtile)
#pragma unroll
        for (int r = 0; r < 16; ++r) pr[r] = __builtin_exp2f(pr[r] - m_new);
        f32x2_ s2 = pr2[0];
#pragma unroll
        for (int r = 1; r < 8; ++r) s2 += pr2[r];
        psum = s2.x + s2.y;
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

      // ---- P^T fragments: pack to bf16 pairs, lazy half-exchange ----
      uint32_t W[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // hardware packed convert (no builtin on gfx950 — inline asm);
        // dtype-correct: f16 kernels must pack f16 fragments (this was
        // v_cvt_pk_bf16_f32 unconditionally — latent fp16 numerics bug
        // caught by test_prefill_fp16)
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
      // NOTE: shfl must run on ALL lanes (uniform control flow) — select after.
      // half-exchange via v_permlane32_swap (guide T12): each call yields
      // BOTH fragments of a (Wlo, Whi) pair — 4 VALU ops replace the
      // former 8 ds_bpermute + 8 selects (verified equal to the select
      // table: b0[0]=concat(W0.lo,W2.lo), b0[2]=concat(W0.hi,W2.hi), ...)
      uint32_t b0[4], b1[4];
      permlane32_pair(W[0], W[2], b0[0], b0[2]);
      permlane32_pair(W[1], W[3], b0[1], b0[3]);
      permlane32_pair(W[4], W[6], b1[0], b1[2]);
      permlane32_pair(W[5], W[7], b1[1], b1[3]);

      // ---- O^T += V^T * P^T: V^T fragments via ds_read_b64_tr_b16 (HW
      // 4x16 transpose read — guide T10); 4 reads + one wait per d-tile ----
      {
        const uint32_t vbase = (uint32_t)(uintptr_t)Vs(buf) + (uint32_t)(lane & 15) * 8;
        const uint32_t tdsel = ((lane >> 4) & 1);
        const int kvb = kt * 32 + khalf;
#pragma unroll
        for (int i = 0; i < DT; ++i) {
          uint32_t a00 = vbase + (((kvb >> 2)) * (HDVO / 16) + i * 2 + tdsel) *
                                     (VTILE_STRIDE * 2);
          b16x4 r00 = ds_read_tr16(a00);
          b16x4 r01 = ds_read_tr16(a00 + (HDVO / 16) * (VTILE_STRIDE * 2));
          uint32_t a10 = a00 + 4 * (HDVO / 16) * (VTILE_STRIDE * 2);
          b16x4 r10 = ds_read_tr16(a10);
          b16x4 r11 = ds_read_tr16(a10 + (HDVO / 16) * (VTILE_STRIDE * 2));
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
          union {
            b16x4 h[2];
            frag f;
          } u0, u1;
          u0.h[0] = r00;
          u0.h[1] = r01;
          u1.h[0] = r10;
          u1.h[1] = r11;
          acc_o[i] = mfma_ab_frag<T>::mma32(u0.f, *reinterpret_cast<frag*>(b0),
                                            acc_o[i]);
          acc_o[i] = mfma_ab_frag<T>::mma32(u1.f, *reinterpret_cast<frag*>(b1),
                                            acc_o[i]);
        }
      }
    }
    }
    if (have_next) stage_write(buf ^ 1);
    __syncthreads();  // one barrier: this tile's reads AND the next tile's
                      // writes (to the other buffer) are both complete
    buf ^= 1;
  }
  prof_event(p.prof_buf, 1, ProfType::kEnd);

  // ---- epilogue: normalize and write O (transpose from O^T frags) ----
  float d_full = xhalf_sum(d_run);
  float inv_d = d_full > 0.f ? 1.f / d_full : 0.f;
  if (row_valid) {
    const int qh = (int)(kv_head * group + my_g_u);
    if (chunk_in >= 0) {
      // split-KV: normalized f32 partial + base-2 lse into the merge slots
      const int slot = p.req_slot_base[req] + my_qpos * n_chunks + chunk;
      float* vout = p.tmp_v + ((int64_t)slot * p.num_qo_heads + qh) * HDVO;
#pragma unroll
      for (int i = 0; i < DT; ++i) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int d = i * 32 + mfma32_cd_row(r, lane);
          vout[d] = acc_o[i][r] * inv_d;
        }
      }
      if ((lane >> 5) == 0) {
        p.tmp_s[(int64_t)slot * p.num_qo_heads + qh] =
            d_full > 0.f ? m_run + __builtin_log2f(d_full) : -INFINITY;
      }
    } else {
      T* optr = (T*)p.out + (int64_t)(qo_begin + my_qpos) * p.o_stride_n +
                (int64_t)qh * p.o_stride_h;
#pragma unroll
      for (int i = 0; i < DT; ++i) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int d = i * 32 + mfma32_cd_row(r, lane);
          optr[d] = from_f32<T>(acc_o[i][r] * inv_d);
        }
      }
      if (p.lse && (lane >> 5) == 0) {
        float l2 = d_full > 0.f ? m_run + __builtin_log2f(d_full) : -INFINITY;
        p.lse[(int64_t)(qo_begin + my_qpos) * p.num_qo_heads + qh] = l2;
      }
    }
  }
  prof_event(p.prof_buf, 0, ProfType::kEnd);
}


}  // namespace fi
