// Sorting-free sampling suite for gfx950. Functional parity with reference
// include/flashinfer/sampling.cuh (OnlineSoftmaxFusedKernel:306,
// SamplingFromProbKernel:785, TopK/TopP/MinP/TopKTopP rejection kernels:
// 849-1202, renorm kernels:1672/1845, ChainSpeculativeSampling:1869), written
// for wave64:
//  * one 512-thread (8-wave) workgroup per batch row; the vocab is streamed
//    in ordered [thread x VEC] chunks.
//  * inverse-CDF walks use a deterministic fixed-tree block scan (wave
//    shfl-scan + LDS wave-total scan) — bitwise reproducible run to run.
//  * rejection sampling (no sort): candidate drawn by inverse-CDF over the
//    pivot-masked distribution, then accepted iff it belongs to the target
//    set (count/top-mass test); otherwise the candidate's prob becomes the
//    new pivot. Uniform randoms [rounds] are drawn by the host from the torch
//    generator.
//  * renorm / mask ops find the k-th (or top-p) threshold with a histogram-
//    free bit-monotonic binary search over float ordering.
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int SB = 512;           // threads per sampling block
constexpr int SW = SB / kWaveSize;  // waves
constexpr int SVEC = 4;           // f32 per thread per chunk
constexpr int CHUNK = SB * SVEC;  // vocab elements per chunk pass


// ---- block reduction helpers (deterministic fixed order) ----
__device__ __forceinline__ float block_sum(float x, float* smem) {
  x = wave_reduce_sum<kWaveSize>(x);
  int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
  if (l == 0) smem[w] = x;
  __syncthreads();
  float r = 0.f;
  if (threadIdx.x == 0) {
#pragma unroll
    for (int i = 0; i < SW; ++i) r += smem[i];
    smem[SW] = r;
  }
  __syncthreads();
  r = smem[SW];
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_max(float x, float* smem) {
  x = wave_reduce_max<kWaveSize>(x);
  int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
  if (l == 0) smem[w] = x;
  __syncthreads();
  float r = -INFINITY;
  if (threadIdx.x == 0) {
#pragma unroll
    for (int i = 0; i < SW; ++i) r = fmaxf(r, smem[i]);
    smem[SW] = r;
  }
  __syncthreads();
  r = smem[SW];
  __syncthreads();
  return r;
}

// exclusive prefix of per-thread totals (deterministic): returns prefix, and
// writes block total to *total.
__device__ __forceinline__ float block_exscan(float x, float* smem, float* total) {
  int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
  // wave inclusive scan
  float inc = x;
#pragma unroll
  for (int off = 1; off < kWaveSize; off <<= 1) {
    float y = __shfl_up(inc, off, 64);
    if (l >= off) inc += y;
  }
  if (l == kWaveSize - 1) smem[w] = inc;
  __syncthreads();
  float wave_off = 0.f, tot = 0.f;
#pragma unroll
  for (int i = 0; i < SW; ++i) {
    if (i < w) wave_off += smem[i];
    tot += smem[i];
  }
  __syncthreads();
  *total = tot;
  return wave_off + (inc - x);
}

// ---------------- softmax (temperature-fused, online) ----------------
__global__ void softmax_kernel(SamplingParams p) {
  __shared__ float smem[SW + 1];
  for (int row = blockIdx.x; row < p.rows; row += gridDim.x) {
    const float* in = p.probs + (uint64_t)row * p.stride_row;
    float* out = p.out_probs + (uint64_t)row * p.vocab;
    float inv_t = p.temperature > 0.f ? 1.f / p.temperature : 1.f;
    float m = -INFINITY, d = 0.f;
    for (int i = threadIdx.x; i < p.vocab; i += SB) {
      float v = in[i] * inv_t;
      float m_new = fmaxf(m, v);
      d = d * __builtin_expf(m - m_new) + __builtin_expf(v - m_new);
      m = m_new;
    }
    // merge per-thread online states
    float gm = block_max(m, smem);
    d = (m == -INFINITY) ? 0.f : d * __builtin_expf(m - gm);
    float gd = block_sum(d, smem);
    float inv_d = gd > 0.f ? 1.f / gd : 0.f;
    for (int i = threadIdx.x; i < p.vocab; i += SB) {
      out[i] = __builtin_expf(in[i] * inv_t - gm) * inv_d;
    }
    __syncthreads();
  }
}

// ---------------- generic rejection sampler ----------------
// MODE: 0 = plain (no constraint), 1 = top-k, 2 = top-p, 3 = top-k AND top-p,
//       4 = min-p (single round: mask p < min_p * max_p)
// vocab cap for the LDS chunk-sum hierarchy (512 chunks x 2048 = 1M ids;
// largest real vocabs are ~256k). The host dispatch guards this.
constexpr int kMaxSampChunks = 512;

template <int MODE, bool FROM_LOGITS>
__global__ void sampling_kernel(SamplingParams p) {
  __shared__ float smem[SW + 1];
  __shared__ float s_csum[kMaxSampChunks];
  __shared__ float s_scalar;
  __shared__ int s_cand;
  for (int b = blockIdx.x; b < p.rows; b += gridDim.x) {
    int row = p.row_indices ? p.row_indices[b] : b;
    const float* in = p.probs + (uint64_t)row * p.stride_row;
    int k = MODE == 1 || MODE == 3
                ? (p.top_k ? p.top_k[b] : p.scalar_k)
                : 0;
    float pp = MODE == 2 || MODE == 3 ? (p.top_p ? p.top_p[b] : p.scalar_p) : 0.f;
    if (k <= 0) k = p.vocab;

    // logits mode: one pass to get (m, d) for softmax normalization
    float lm = 0.f, ld = 1.f;
    if constexpr (FROM_LOGITS) {
      float m = -INFINITY, d = 0.f;
      for (int i = threadIdx.x; i < p.vocab; i += SB) {
        float v = in[i];
        float m_new = fmaxf(m, v);
        d = d * __builtin_expf(m - m_new) + __builtin_expf(v - m_new);
        m = m_new;
      }
      float gm = block_max(m, smem);
      d = (m == -INFINITY) ? 0.f : d * __builtin_expf(m - gm);
      lm = gm;
      ld = block_sum(d, smem);
    }
    // prob accessor
    auto P = [&](int i) -> float {
      float v = in[i];
      if constexpr (FROM_LOGITS) return __builtin_expf(v - lm) / ld;
      return v;
    };

    float pivot = -1.f;  // accept probs strictly > pivot
    if constexpr (MODE == 4) {
      // min-p: pivot = min_p * max_prob (single round, no rejection)
      float mx = -INFINITY;
      for (int i = threadIdx.x; i < p.vocab; i += SB) mx = fmaxf(mx, P(i));
      mx = block_max(mx, smem);
      float mp = p.top_p ? p.top_p[b] : p.min_p;
      pivot = mx * mp * 0.999999f;  // epsilon so the max itself stays included
      if (pivot <= 0.f) pivot = -1.f;
    }

    // ---- per-chunk partial-sum hierarchy (r02 roadmap item 7) ----
    // One full-vocab pass builds s_csum[c] = sum of probs > pivot in chunk
    // c; the inverse-CDF walk then touches ONE chunk (selected by a serial
    // deterministic scan of s_csum). Each rejection round fuses the next
    // pivot's rebuild into the acceptance pass, so a round costs ~1 vocab
    // pass instead of the previous ~2.5 (pass1 + half-vocab walk + accept).
    const int nchunks = (p.vocab + CHUNK - 1) / CHUNK;
    const int wv = threadIdx.x / kWaveSize, lane = threadIdx.x % kWaveSize;
    {
      for (int c = wv; c < nchunks; c += SW) {
        float sch = 0.f;
        const int base = c * CHUNK;
        const int lim = base + CHUNK < p.vocab ? base + CHUNK : p.vocab;
        for (int i = base + lane; i < lim; i += kWaveSize) {
          float v = P(i);
          if (v > pivot) sch += v;
        }
        sch = wave_reduce_sum<kWaveSize>(sch);
        if (lane == 0) s_csum[c] = sch;
      }
      __syncthreads();
    }

    int cand = -1;
    for (int round = 0; round < p.rounds; ++round) {
      float u = p.uniforms[(uint64_t)b * p.rounds + round];
      // chunk select: serial scan of the per-chunk sums (deterministic,
      // <= kMaxSampChunks dependent adds — negligible next to a pass)
      if (threadIdx.x == 0) {
        float total = 0.f;
        for (int c = 0; c < nchunks; ++c) total += s_csum[c];
        int cstar = -1;
        float run = 0.f;
        if (total > 0.f) {
          float target = u * total;
          for (int c = 0; c < nchunks; ++c) {
            float cs = s_csum[c];
            if (run + cs > target && cs > 0.f) { cstar = c; break; }
            run += cs;
          }
          if (cstar < 0) {
            // rounding tail: last chunk with mass
            run = 0.f;
            for (int c = 0; c < nchunks; ++c) {
              if (s_csum[c] > 0.f) cstar = c;
            }
            for (int c = 0; c < cstar; ++c) run += s_csum[c];
            target = run + s_csum[cstar];  // clamp into the chunk
          }
          s_scalar = target - run;  // in-chunk target mass
        }
        s_cand = cstar;
      }
      __syncthreads();
      const int cstar = s_cand;
      if (cstar < 0) break;  // nothing above pivot (keep last cand)
      const float trem = s_scalar;
      __syncthreads();
      // in-chunk inverse-CDF walk (one chunk == one CHUNK-wide block scan)
      const int i0 = cstar * CHUNK + threadIdx.x * SVEC;
      float vals[SVEC];
      float th_sum = 0.f;
#pragma unroll
      for (int j = 0; j < SVEC; ++j) {
        int i = i0 + j;
        float v = (i < p.vocab) ? P(i) : 0.f;
        vals[j] = (v > pivot) ? v : 0.f;
        th_sum += vals[j];
      }
      float ctot;
      float loc = block_exscan(th_sum, smem, &ctot);
      int my = INT_MAX, last_valid = -1;
#pragma unroll
      for (int j = 0; j < SVEC; ++j) {
        if (vals[j] > 0.f) {
          if (my == INT_MAX && loc + vals[j] > trem) my = i0 + j;
          loc += vals[j];
          last_valid = i0 + j;
        }
      }
      if (threadIdx.x == 0) s_cand = INT_MAX;
      __syncthreads();
      if (my != INT_MAX) atomicMin(&s_cand, my);
      __syncthreads();
      int found = s_cand;
      if (found == INT_MAX) {
        // rounding edge: target beyond the chunk's walked mass — take the
        // chunk's last valid element
        if (threadIdx.x == 0) s_cand = -1;
        __syncthreads();
        if (last_valid >= 0) atomicMax(&s_cand, last_valid);
        __syncthreads();
        found = s_cand;
        if (found < 0) break;
      }
      cand = found;
      if constexpr (MODE == 0 || MODE == 4) break;  // no rejection test
      // acceptance test FUSED with the csum rebuild for pivot=pc: if this
      // candidate is rejected the next round's hierarchy is already built
      float pc = P(cand);
      float gt_sum = 0.f;
      float gt_cnt = 0.f;
      for (int c = wv; c < nchunks; c += SW) {
        float sch = 0.f;
        const int base = c * CHUNK;
        const int lim = base + CHUNK < p.vocab ? base + CHUNK : p.vocab;
        for (int i = base + lane; i < lim; i += kWaveSize) {
          float v = P(i);
          if (v > pc) {
            sch += v;
            gt_cnt += 1.f;
          }
        }
        gt_sum += sch;
        sch = wave_reduce_sum<kWaveSize>(sch);
        if (lane == 0) s_csum[c] = sch;
      }
      float g_sum = block_sum(gt_sum, smem);
      float g_cnt = block_sum(gt_cnt, smem);
      bool ok = true;
      if constexpr (MODE == 1 || MODE == 3) ok &= (g_cnt < k);
      if constexpr (MODE == 2 || MODE == 3) ok &= (g_sum < pp);
      if (ok) break;
      pivot = pc;  // reject: tighten (s_csum already rebuilt for pc)
    }
    if (threadIdx.x == 0) p.out_ids[b] = cand < 0 ? 0 : cand;
    __syncthreads();
  }
}

// ---------------- renorm / mask via threshold search ----------------
// Monotonic uint mapping of floats (handles negatives) for binary search.
__device__ __forceinline__ uint32_t f2u(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float u2f(uint32_t u) {
  return __uint_as_float((u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u);
}

// WHICH: 0 = top_k_renorm_probs, 1 = top_p_renorm_probs, 2 = top_k_mask_logits
// The k-th/top-p threshold tau is found EXACTLY (full 32-bit float ordering)
// by a 3-level radix histogram over the monotonic-uint space (11+11+10
// bits): 3 vocab passes instead of the former 24-iteration binary search
// (~24 passes). Counts use integer LDS atomics (order-independent, so
// top-k results stay bitwise deterministic); per-bin float sums (top-p bin
// selection only) use float atomics — the final renorm sum is recomputed
// in the fixed block_sum order.
// top-p bin masses use FIXED-POINT u64 LDS atomics (probs in [0,1] scaled
// by 2^40): integer adds are order-free, so the selected threshold — and
// therefore the whole output — is bitwise deterministic run to run, which
// float atomicAdd is not (quantization error <= vocab * 2^-40 ~ 1e-7, the
// same class as the float rounding it replaces).
constexpr float kPScale = 1099511627776.f;  // 2^40

template <int WHICH>
__global__ void renorm_kernel(SamplingParams p) {
  __shared__ float smem[SW + 1];
  __shared__ uint32_t s_hcnt[2048];
  __shared__ uint64_t s_hsum[2048];
  __shared__ uint32_t s_sel;
  __shared__ float s_carry;
  __shared__ uint64_t s_carry64;
  for (int b = blockIdx.x; b < p.rows; b += gridDim.x) {
    const float* in = p.probs + (uint64_t)b * p.stride_row;
    float* out = p.out_probs + (uint64_t)b * p.vocab;
    int k = p.top_k ? p.top_k[b] : p.scalar_k;
    float pp = p.top_p ? p.top_p[b] : p.scalar_p;
    if (WHICH != 1 && (k <= 0 || k >= p.vocab)) {
      // no-op (copy)
      for (int i = threadIdx.x; i < p.vocab; i += SB) out[i] = in[i];
      __syncthreads();
      continue;
    }
    const uint64_t ppq = (uint64_t)((double)pp * (double)kPScale);
    uint32_t prefix = 0;      // matched high bits of tau's monotonic uint
    float cum_cnt = 0.f;      // count strictly above prefix region
    uint64_t cum_q = 0;       // fixed-point mass strictly above
    constexpr int kShift[3] = {21, 10, 0};
    constexpr int kBits[3] = {11, 11, 10};
    for (int lvl = 0; lvl < 3; ++lvl) {
      const int nb = 1 << kBits[lvl];
      const int hi_shift = kShift[lvl] + kBits[lvl];  // bits already matched
      for (int i = threadIdx.x; i < nb; i += SB) {
        s_hcnt[i] = 0;
        if constexpr (WHICH == 1) s_hsum[i] = 0;
      }
      __syncthreads();
      for (int i = threadIdx.x; i < p.vocab; i += SB) {
        float v = in[i];
        uint32_t u = f2u(v);
        if (lvl == 0 || (u >> hi_shift) == prefix) {
          int bin = (u >> kShift[lvl]) & (nb - 1);
          // top-p selects on mass only; skip the count atomic there
          if constexpr (WHICH == 1) {
            atomicAdd((unsigned long long*)&s_hsum[bin],
                      (unsigned long long)(fmaxf(v, 0.f) * kPScale));
          } else {
            atomicAdd(&s_hcnt[bin], 1u);
          }
        }
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        int chosen = -1;
        float cc = cum_cnt;
        uint64_t cq = cum_q;
        for (int bin = nb - 1; bin >= 0; --bin) {
          bool present = (WHICH == 1) ? (s_hsum[bin] != 0) : (s_hcnt[bin] > 0);
          if (present) {
            bool crossed = (WHICH == 1) ? (cq + s_hsum[bin] >= ppq)
                                        : (cc + (float)s_hcnt[bin] >= (float)k);
            if (crossed) { chosen = bin; break; }
          }
          if constexpr (WHICH == 1) cq += s_hsum[bin];
          else cc += (float)s_hcnt[bin];
        }
        if (chosen < 0) {
          // never crossed (rounding tail / pp > total): lowest present bin
          cc = cum_cnt;
          cq = cum_q;
          for (int bin = nb - 1; bin >= 0; --bin) {
            bool present = (WHICH == 1) ? (s_hsum[bin] != 0) : (s_hcnt[bin] > 0);
            if (present) chosen = bin;
          }
          if (chosen < 0) chosen = 0;
          for (int bin = nb - 1; bin > chosen; --bin) {
            if constexpr (WHICH == 1) cq += s_hsum[bin];
            else cc += (float)s_hcnt[bin];
          }
        }
        s_sel = (uint32_t)chosen;
        s_carry = cc;
        s_carry64 = cq;
      }
      __syncthreads();
      prefix = (prefix << kBits[lvl]) | s_sel;
      cum_cnt = s_carry;
      cum_q = s_carry64;
      __syncthreads();
    }
    float tau = u2f(prefix);  // exact k-th / top-p boundary value
    // renormalize / mask
    float ssum = 0.f;
    if (WHICH != 2) {
      for (int i = threadIdx.x; i < p.vocab; i += SB) {
        float v = in[i];
        if (v >= tau) ssum += v;
      }
      ssum = block_sum(ssum, smem);
    }
    float inv = ssum > 0.f ? 1.f / ssum : 0.f;
    for (int i = threadIdx.x; i < p.vocab; i += SB) {
      float v = in[i];
      if (WHICH == 2) out[i] = (v >= tau) ? v : -INFINITY;
      else out[i] = (v >= tau) ? v * inv : 0.f;
    }
    __syncthreads();
  }
}

// ---------------- chain speculative sampling ----------------

__global__ void chain_spec_kernel(SpecParams sp) {
  __shared__ float smem[SW + 1];
  __shared__ int s_cand;
  int b = blockIdx.x;
  if (b >= sp.B) return;
  int V = sp.vocab;
  int pos = 0;
  bool all_accepted = true;
  for (; pos < sp.n; ++pos) {
    int t = sp.draft_ids[(uint64_t)b * sp.n + pos];
    float pd = sp.draft_probs[((uint64_t)b * sp.n + pos) * V + t];
    float pt = sp.target_probs[((uint64_t)b * (sp.n + 1) + pos) * V + t];
    float u = sp.uniforms[(uint64_t)b * (sp.n + 1) + pos];
    bool accept = u * pd <= pt;  // u <= pt/pd (pd>0)
    if (accept) {
      if (threadIdx.x == 0) sp.out_ids[(uint64_t)b * (sp.n + 1) + pos] = t;
    } else {
      all_accepted = false;
      break;
    }
  }
  if (threadIdx.x == 0) {
    // accepted_num keeps running the accept test past the first rejection
    // (reference sampling.cuh:1906-1920 contract: accepted >= emitted);
    // emitted_num counts only the prefix actually emitted.
    int accepted = pos;
    for (int i = pos; i < sp.n; ++i) {
      int t = sp.draft_ids[(uint64_t)b * sp.n + i];
      float pd = sp.draft_probs[((uint64_t)b * sp.n + i) * V + t];
      float pt = sp.target_probs[((uint64_t)b * (sp.n + 1) + i) * V + t];
      float u = sp.uniforms[(uint64_t)b * (sp.n + 1) + i];
      if (u * pd <= pt) ++accepted;
    }
    atomicAdd(sp.accepted_num + b, accepted);
    atomicAdd(sp.emitted_num + b, pos);
  }
  // sample one more token: from residual at `pos` (rejected) or from the
  // last target distribution (all accepted)
  const float* tp = sp.target_probs + ((uint64_t)b * (sp.n + 1) + pos) * V;
  const float* dp = all_accepted ? nullptr
                                 : sp.draft_probs + ((uint64_t)b * sp.n + pos) * V;
  auto R = [&](int i) -> float {
    float v = tp[i];
    if (dp) v = fmaxf(v - dp[i], 0.f);
    return v;
  };
  float tsum = 0.f;
  for (int i = threadIdx.x; i < V; i += SB) tsum += R(i);
  float total = block_sum(tsum, smem);
  float u = sp.uniforms[(uint64_t)b * (sp.n + 1) + sp.n];
  float target = u * total;
  int found = INT_MAX;
  float running = 0.f;
  int last_valid = -1;
  for (int base = 0; base < V; base += CHUNK) {
    float vals[SVEC];
    int i0 = base + threadIdx.x * SVEC;
    float th_sum = 0.f;
#pragma unroll
    for (int j = 0; j < SVEC; ++j) {
      int i = i0 + j;
      vals[j] = (i < V) ? R(i) : 0.f;
      th_sum += vals[j];
    }
    float ctot;
    float prefix = block_exscan(th_sum, smem, &ctot);
    if (running + ctot > target) {
      float loc = running + prefix;
      int my = INT_MAX;
#pragma unroll
      for (int j = 0; j < SVEC; ++j) {
        if (vals[j] > 0.f) {
          if (my == INT_MAX && loc + vals[j] > target) my = i0 + j;
          loc += vals[j];
        }
      }
      if (threadIdx.x == 0) s_cand = INT_MAX;
      __syncthreads();
      if (my != INT_MAX) atomicMin(&s_cand, my);
      __syncthreads();
      found = s_cand;
      if (found != INT_MAX) break;
      running += ctot;
    } else {
      running += ctot;
    }
#pragma unroll
    for (int j = 0; j < SVEC; ++j)
      if (vals[j] > 0.f) last_valid = i0 + j;
  }
  if (found == INT_MAX) {
    if (threadIdx.x == 0) s_cand = -1;
    __syncthreads();
    if (last_valid >= 0) atomicMax(&s_cand, last_valid);
    __syncthreads();
    found = s_cand < 0 ? 0 : s_cand;
  }
  if (threadIdx.x == 0) {
    sp.out_ids[(uint64_t)b * (sp.n + 1) + pos] = found;
    // fill the rest with -1
    for (int i = pos + 1; i <= sp.n; ++i)
      sp.out_ids[(uint64_t)b * (sp.n + 1) + i] = -1;
  }
}

}  // namespace fi

extern "C" hipError_t fi_softmax(fi::SamplingParams* p, hipStream_t stream) {
  int grid = p->rows < 1024 ? p->rows : 1024;
  hipLaunchKernelGGL(fi::softmax_kernel, dim3(grid), dim3(fi::SB), 0, stream, *p);
  return hipGetLastError();
}

// mode: 0 plain, 1 topk, 2 topp, 3 topk+topp, 4 minp; from_logits bool
extern "C" hipError_t fi_sampling(int mode, int from_logits, fi::SamplingParams* p,
                                  hipStream_t stream) {
  if (p->vocab > fi::kMaxSampChunks * fi::CHUNK) return hipErrorInvalidValue;
  int grid = p->rows < 1024 ? p->rows : 1024;
  dim3 g(grid), blk(fi::SB);
#define LAUNCH_S(M, L) \
  hipLaunchKernelGGL((fi::sampling_kernel<M, L>), g, blk, 0, stream, *p)
  switch (mode * 2 + (from_logits ? 1 : 0)) {
    case 0: LAUNCH_S(0, false); break;
    case 1: LAUNCH_S(0, true); break;
    case 2: LAUNCH_S(1, false); break;
    case 3: LAUNCH_S(1, true); break;
    case 4: LAUNCH_S(2, false); break;
    case 5: LAUNCH_S(2, true); break;
    case 6: LAUNCH_S(3, false); break;
    case 7: LAUNCH_S(3, true); break;
    case 8: LAUNCH_S(4, false); break;
    case 9: LAUNCH_S(4, true); break;
    default: return hipErrorInvalidValue;
  }
#undef LAUNCH_S
  return hipGetLastError();
}

// which: 0 top_k_renorm_probs, 1 top_p_renorm_probs, 2 top_k_mask_logits
extern "C" hipError_t fi_renorm(int which, fi::SamplingParams* p, hipStream_t stream) {
  int grid = p->rows < 1024 ? p->rows : 1024;
  dim3 g(grid), blk(fi::SB);
  switch (which) {
    case 0: hipLaunchKernelGGL((fi::renorm_kernel<0>), g, blk, 0, stream, *p); break;
    case 1: hipLaunchKernelGGL((fi::renorm_kernel<1>), g, blk, 0, stream, *p); break;
    case 2: hipLaunchKernelGGL((fi::renorm_kernel<2>), g, blk, 0, stream, *p); break;
    default: return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

extern "C" hipError_t fi_chain_speculative(fi::SpecParams* sp, hipStream_t stream) {
  hipLaunchKernelGGL(fi::chain_spec_kernel, dim3(sp->B), dim3(fi::SB), 0, stream, *sp);
  return hipGetLastError();
}
