// LSE-weighted attention-state merge kernels — the foundation of split-KV,
// cascade (shared-prefix) attention and ring attention. Parity with reference
// include/flashinfer/attention/cascade.cuh (MergeState:45, MergeStateInPlace:90,
// MergeStates:183, VariableLengthMergeStates:374) re-derived for wave64.
//
// Convention: public s/lse values are BASE-2 log-sum-exp (reference-compatible,
// flashinfer trace templates treat s as log2 scale); internal state math is
// natural-log.
// All merging math is f32; inputs/outputs may be bf16/fp16/f32.
#include "fi/common.hpp"
#include "fi/state.hpp"
#include "fi/vec.hpp"

namespace fi {

// One wave per (pos, head); each lane owns a VEC-wide slice of head_dim.
// Merges `count` partial states read via `chunk_stride` starting at base
// index `first`. Handles count==0 by writing zeros / -inf lse.
template <typename TIn, typename TOut, int VEC>
__global__ void merge_states_kernel(const TIn* __restrict__ v_in,
                                    const float* __restrict__ s_in,
                                    TOut* __restrict__ v_out, float* __restrict__ s_out,
                                    const int32_t* __restrict__ merge_indptr,
                                    int64_t uniform_count, int64_t num_pos, int num_heads,
                                    int head_dim) {
  int64_t wave_id = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWaveSize;
  int lane = threadIdx.x % kWaveSize;
  int64_t total = num_pos * num_heads;
  if (wave_id >= total) return;
  int head = (int)(wave_id % num_heads);
  int64_t pos = wave_id / num_heads;

  int64_t first, count;
  if (merge_indptr) {
    first = merge_indptr[pos];
    count = merge_indptr[pos + 1] - first;
  } else {
    first = pos * uniform_count;
    count = uniform_count;
  }

  int d0 = lane * VEC;
  state_t<VEC> st;
  st.init();
  if (d0 < head_dim) {
    for (int64_t c = 0; c < count; ++c) {
      int64_t idx = first + c;  // chunk row index
      float s = s_in[idx * num_heads + head] * 0.6931471805599453f;  // log2 -> ln
      float vv[VEC];
      vec_t<TIn, VEC> vload;
      vload.load(v_in + (idx * num_heads + head) * (int64_t)head_dim + d0);
#pragma unroll
      for (int j = 0; j < VEC; ++j) vv[j] = vload.get(j);
      // merge a normalized partial: o_norm with lse s  ==  state(m=s, d=1, o=o_norm)
      st.merge(vv, s, 1.f);
    }
    st.normalize();
    vec_t<TOut, VEC> vout;
#pragma unroll
    for (int j = 0; j < VEC; ++j) vout.set(j, st.o[j]);
    vout.store(v_out + (pos * num_heads + head) * (int64_t)head_dim + d0);
  }
  if (lane == 0 && s_out) s_out[pos * num_heads + head] = st.lse() * 1.4426950408889634f;
}

// Two-way in-place merge: (v, s) <- merge((v, s), (v_other, s_other)).
template <typename T, int VEC>
__global__ void merge_state_in_place_kernel(T* __restrict__ v, float* __restrict__ s,
                                            const T* __restrict__ v_other,
                                            const float* __restrict__ s_other,
                                            int64_t num_pos, int num_heads, int head_dim,
                                            const uint8_t* __restrict__ mask) {
  int64_t wave_id = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWaveSize;
  int lane = threadIdx.x % kWaveSize;
  int64_t total = num_pos * num_heads;
  if (wave_id >= total) return;
  int head = (int)(wave_id % num_heads);
  int64_t pos = wave_id / num_heads;
  if (mask && !mask[pos]) return;

  int d0 = lane * VEC;
  if (d0 >= head_dim) return;
  int64_t base = (pos * num_heads + head) * (int64_t)head_dim + d0;
  float s_a = s[pos * num_heads + head] * 0.6931471805599453f;
  float s_b = s_other[pos * num_heads + head] * 0.6931471805599453f;
  state_t<VEC> st;
  st.init();
  float va[VEC], vb[VEC];
  vec_t<T, VEC> la, lb;
  la.load(v + base);
  lb.load(v_other + base);
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    va[j] = la.get(j);
    vb[j] = lb.get(j);
  }
  st.merge(va, s_a, 1.f);
  st.merge(vb, s_b, 1.f);
  st.normalize();
  vec_t<T, VEC> out;
#pragma unroll
  for (int j = 0; j < VEC; ++j) out.set(j, st.o[j]);
  out.store(v + base);
  if (lane == 0) s[pos * num_heads + head] = st.lse() * 1.4426950408889634f;
}

}  // namespace fi

// state merge entry points. dtype codes: 0 bf16, 1 fp16, 2 fp32 (applies to
// both in and out; mixed f32-in/bf16-out used by split-kv goes through
// in_dtype=2/out via out_dtype).
extern "C" hipError_t fi_merge_states(int in_dtype, int out_dtype, const void* v_in,
                                      const float* s_in, void* v_out, float* s_out,
                                      const int32_t* merge_indptr, int64_t uniform_count,
                                      int64_t num_pos, int num_heads, int head_dim,
                                      hipStream_t stream) {
  int vec = fi::ceil_div(head_dim, fi::kWaveSize);
  if (vec > 8) return hipErrorInvalidValue;
  int64_t waves = num_pos * num_heads;
  int waves_per_block = 4;
  int64_t blocks = (waves + waves_per_block - 1) / waves_per_block;
  dim3 g((uint32_t)blocks), blk(waves_per_block * fi::kWaveSize);

#define LAUNCH_M(TI, TO, V)                                                           \
  hipLaunchKernelGGL((fi::merge_states_kernel<TI, TO, V>), g, blk, 0, stream,         \
                     (const TI*)v_in, s_in, (TO*)v_out, s_out, merge_indptr,          \
                     uniform_count, num_pos, num_heads, head_dim)
#define DISPATCH_V(TI, TO)                                                            \
  do {                                                                                \
    switch (vec) {                                                                    \
      case 1: LAUNCH_M(TI, TO, 1); break;                                             \
      case 2: LAUNCH_M(TI, TO, 2); break;                                             \
      case 4: LAUNCH_M(TI, TO, 4); break;                                             \
      case 8: LAUNCH_M(TI, TO, 8); break;                                             \
      default: {                                                                      \
        int v2 = vec <= 2 ? 2 : (vec <= 4 ? 4 : 8);                                   \
        if (v2 == 2) LAUNCH_M(TI, TO, 2);                                             \
        else if (v2 == 4) LAUNCH_M(TI, TO, 4);                                        \
        else LAUNCH_M(TI, TO, 8);                                                     \
      }                                                                               \
    }                                                                                 \
  } while (0)

  if (in_dtype == 2 && out_dtype == 2) DISPATCH_V(float, float);
  else if (in_dtype == 2 && out_dtype == 0) DISPATCH_V(float, fi::bf16);
  else if (in_dtype == 2 && out_dtype == 1) DISPATCH_V(float, fi::fp16);
  else if (in_dtype == 0 && out_dtype == 0) DISPATCH_V(fi::bf16, fi::bf16);
  else if (in_dtype == 1 && out_dtype == 1) DISPATCH_V(fi::fp16, fi::fp16);
  else return hipErrorInvalidValue;
#undef DISPATCH_V
#undef LAUNCH_M
  return hipGetLastError();
}

extern "C" hipError_t fi_merge_state_in_place(int dtype, void* v, float* s,
                                              const void* v_other, const float* s_other,
                                              int64_t num_pos, int num_heads, int head_dim,
                                              const uint8_t* mask, hipStream_t stream) {
  int vec = fi::ceil_div(head_dim, fi::kWaveSize);
  if (vec > 8) return hipErrorInvalidValue;
  int64_t waves = num_pos * num_heads;
  int waves_per_block = 4;
  int64_t blocks = (waves + waves_per_block - 1) / waves_per_block;
  dim3 g((uint32_t)blocks), blk(waves_per_block * fi::kWaveSize);
#define LAUNCH_I(T, V)                                                              \
  hipLaunchKernelGGL((fi::merge_state_in_place_kernel<T, V>), g, blk, 0, stream,    \
                     (T*)v, s, (const T*)v_other, s_other, num_pos, num_heads,      \
                     head_dim, mask)
#define DISPATCH_I(T)                                                               \
  do {                                                                              \
    if (vec == 1) LAUNCH_I(T, 1);                                                   \
    else if (vec == 2) LAUNCH_I(T, 2);                                              \
    else if (vec <= 4) LAUNCH_I(T, 4);                                              \
    else LAUNCH_I(T, 8);                                                            \
  } while (0)
  switch (dtype) {
    case 0: DISPATCH_I(fi::bf16); break;
    case 1: DISPATCH_I(fi::fp16); break;
    case 2: DISPATCH_I(float); break;
    default: return hipErrorInvalidValue;
  }
#undef DISPATCH_I
#undef LAUNCH_I
  return hipGetLastError();
}
