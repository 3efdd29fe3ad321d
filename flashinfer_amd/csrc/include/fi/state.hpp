// Online-softmax attention state (flash-attention running (m, d, o) triple)
// and the LSE-weighted merge primitive used by split-KV, cascade attention
// and ring attention. Functional equivalent of the reference's
// include/flashinfer/attention/state.cuh + cascade.cuh merge ops, re-derived
// for CDNA4 (wave64) from the math.
#pragma once
#include "fi/common.hpp"

namespace fi {

// Per-thread online softmax state over a vec_size-wide slice of the output.
template <int vec_size>
struct state_t {
  float o[vec_size];  // un-normalized weighted-V accumulator slice
  float m;            // running max of logits
  float d;            // running sum of exp(logit - m)

  __device__ __forceinline__ void init() {
#pragma unroll
    for (int i = 0; i < vec_size; ++i) o[i] = 0.f;
    m = -INFINITY;
    d = 0.f;
  }

  // Fold in one logit `s` with V-slice `v` (already f32).
  __device__ __forceinline__ void push(const float* v, float s) {
    float m_new = fmaxf(m, s);
    if (m_new == -INFINITY) return;  // everything masked so far
    float scale = __builtin_expf(m - m_new);
    float p = __builtin_expf(s - m_new);
    d = d * scale + p;
#pragma unroll
    for (int i = 0; i < vec_size; ++i) o[i] = o[i] * scale + p * v[i];
    m = m_new;
  }

  // Merge another partial state (other.o un-normalized, other.m/d valid).
  __device__ __forceinline__ void merge(const float* o_other, float m_other, float d_other) {
    float m_new = fmaxf(m, m_other);
    if (m_new == -INFINITY) return;  // both sides empty
    float s1 = __builtin_expf(m - m_new);
    float s2 = __builtin_expf(m_other - m_new);
    d = d * s1 + d_other * s2;
#pragma unroll
    for (int i = 0; i < vec_size; ++i) o[i] = o[i] * s1 + o_other[i] * s2;
    m = m_new;
  }

  __device__ __forceinline__ void normalize() {
    float inv_d = (d > 0.f) ? 1.f / d : 0.f;
#pragma unroll
    for (int i = 0; i < vec_size; ++i) o[i] *= inv_d;
  }

  // log-sum-exp of everything pushed so far (base e).
  __device__ __forceinline__ float lse() const {
    return (d > 0.f) ? m + __builtin_logf(d) : -INFINITY;
  }
};

}  // namespace fi
