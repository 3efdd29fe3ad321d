// One-shot custom allreduce over hipIpc-mapped peer buffers (reference
// parity: include/flashinfer/comm/trtllm_allreduce.cuh
// oneShotAllReduceKernel:1166 + fused AR+RMSNorm:394 — re-designed for the
// xGMI point-to-point fabric: every rank reads all peers' buffers directly,
// 7 links x ~153 GB/s, no switch).
//
// Protocol (Lamport-style monotonic sequence flags):
//   host: async-copy input into own buffer's data region (stream-ordered)
//   kernel: block 0 publishes own seq flag (after a system fence);
//           all blocks spin on every peer's flag >= seq (bounded spin —
//           on expiry the kernel aborts and reports, never hangs the GPU);
//           grid-stride f32 sum over all ranks' data -> out.
// Buffer layout per rank: [64B flag slot (uint64 seq)][data ...].
#include "fi/common.hpp"
#include "fi/params.hpp"
#include "fi/vec.hpp"

namespace fi {

constexpr int kARFlagBytes = 64;

__device__ __forceinline__ unsigned long long* ar_flag(unsigned long long buf) {
  return (unsigned long long*)buf;
}

template <typename T>
__device__ __forceinline__ const T* ar_data(const ARParams& p, int r) {
  return (const T*)(p.bufs[r] + p.data_off);
}

// returns false (uniformly across the block) on timeout
__device__ __forceinline__ bool ar_sync(const ARParams& p, int* s_ok) {
  if (threadIdx.x == 0) *s_ok = 1;
  __syncthreads();
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    __threadfence_system();
    atomicExch(ar_flag(p.bufs[p.rank]), p.seq);
  }
  if (threadIdx.x < kMaxRanks && (int)threadIdx.x < p.world) {
    int r = threadIdx.x;
    unsigned long long it = 0;
    while (atomicAdd(ar_flag(p.bufs[r]), 0ull) < p.seq) {
      if (++it > p.spin_limit) {
        atomicExch(p.error_flag, 1);
        *s_ok = 0;
        break;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }
  __syncthreads();
  __threadfence_system();
  return *s_ok != 0;
}

template <typename T>
__global__ void one_shot_ar_kernel(ARParams p, T* __restrict__ out, int64_t numel) {
  __shared__ int s_ok;
  if (!ar_sync(p, &s_ok)) return;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < numel;
       i += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int r = 0; r < p.world; ++r) acc += to_f32<T>(ar_data<T>(p, r)[i]);
    out[i] = from_f32<T>(acc);
  }
}

// Fused AR + residual-add + RMSNorm (reference allreduce_fusion pattern
// kARResidualRMSNorm): residual += allreduce(x); out = rmsnorm(residual)*w.
// One block per row so the row reduction stays in one workgroup.
template <typename T>
__global__ void one_shot_ar_rmsnorm_kernel(ARParams p, T* __restrict__ out,
                                           T* __restrict__ residual,
                                           const T* __restrict__ weight, int rows,
                                           int d, float eps) {
  __shared__ float smem[5];
  __shared__ int s_ok;
  if (!ar_sync(p, &s_ok)) return;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    float ss = 0.f;
    for (int i = threadIdx.x; i < d; i += blockDim.x) {
      int64_t idx = (int64_t)row * d + i;
      float acc = residual ? to_f32<T>(residual[idx]) : 0.f;
      for (int r = 0; r < p.world; ++r) acc += to_f32<T>(ar_data<T>(p, r)[idx]);
      if (residual) residual[idx] = from_f32<T>(acc);
      else out[idx] = from_f32<T>(acc);  // stash pre-norm sum
      ss += acc * acc;
    }
    // block reduce (256 threads = 4 waves)
    ss = wave_reduce_sum<kWaveSize>(ss);
    int w = threadIdx.x / kWaveSize, l = threadIdx.x % kWaveSize;
    if (l == 0) smem[w] = ss;
    __syncthreads();
    if (threadIdx.x == 0) smem[4] = smem[0] + smem[1] + smem[2] + smem[3];
    __syncthreads();
    float rrms = rsqrtf(smem[4] / d + eps);
    for (int i = threadIdx.x; i < d; i += blockDim.x) {
      int64_t idx = (int64_t)row * d + i;
      float v = to_f32<T>(residual ? residual[idx] : out[idx]);
      out[idx] = from_f32<T>(v * rrms * to_f32<T>(weight[i]));
    }
    __syncthreads();
  }
}

}  // namespace fi

extern "C" hipError_t fi_one_shot_ar(int dtype, fi::ARParams* p, void* out,
                                     int64_t numel, hipStream_t stream) {
  int grid = (int)((numel + 255) / 256);
  if (grid > 128) grid = 128;  // small grid: two ranks may share one GPU
  if (grid == 0) grid = 1;
  dim3 g(grid), blk(256);
  switch (dtype) {
    case 0: hipLaunchKernelGGL((fi::one_shot_ar_kernel<fi::bf16>), g, blk, 0, stream, *p, (fi::bf16*)out, numel); break;
    case 1: hipLaunchKernelGGL((fi::one_shot_ar_kernel<fi::fp16>), g, blk, 0, stream, *p, (fi::fp16*)out, numel); break;
    case 2: hipLaunchKernelGGL((fi::one_shot_ar_kernel<float>), g, blk, 0, stream, *p, (float*)out, numel); break;
    default: return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

extern "C" hipError_t fi_one_shot_ar_rmsnorm(int dtype, fi::ARParams* p, void* out,
                                             void* residual, const void* weight,
                                             int rows, int d, float eps,
                                             hipStream_t stream) {
  int grid = rows < 128 ? rows : 128;
  if (grid == 0) grid = 1;
  dim3 g(grid), blk(256);
#define LAR(T)                                                                  \
  hipLaunchKernelGGL((fi::one_shot_ar_rmsnorm_kernel<T>), g, blk, 0, stream,    \
                     *p, (T*)out, (T*)residual, (const T*)weight, rows, d, eps)
  switch (dtype) {
    case 0: LAR(fi::bf16); break;
    case 1: LAR(fi::fp16); break;
    case 2: LAR(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LAR
  return hipGetLastError();
}

// ------------------- two-shot RS + AG allreduce -------------------
// Reference role: trtllm_allreduce.cuh twoShotAllReduceKernel:1279, but
// shaped for xGMI point-to-point: phase 1 (reduce-scatter) has every rank
// PULL its own 1/world shard from all 7 peers concurrently (all links busy,
// total remote traffic ~ numel vs the one-shot's 7x numel); phase 2
// (all-gather) pulls each peer's reduced shard. The phases are two kernel
// launches — the kernel boundary is the device-wide sync that makes the
// shard complete before its flag publishes. Shard slots are double-buffered
// by seq parity like the data slots; flags[1] is the shard-ready sequence.

namespace fi {

// reduce own shard into own shard slot
template <typename T>
__global__ void two_shot_rs_kernel(ARParams p, unsigned long long shard_off,
                                   int64_t numel) {
  __shared__ int s_ok;
  if (!ar_sync(p, &s_ok)) return;
  int64_t per = (numel + p.world - 1) / p.world;
  int64_t s0 = (int64_t)p.rank * per;
  int64_t s1 = s0 + per < numel ? s0 + per : numel;
  T* dst = (T*)(p.bufs[p.rank] + shard_off);
  for (int64_t i = s0 + blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < s1;
       i += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int r = 0; r < p.world; ++r) acc += to_f32<T>(ar_data<T>(p, r)[i]);
    dst[i - s0] = from_f32<T>(acc);
  }
}

// publish shard-ready, wait peers, gather shards into out
template <typename T>
__global__ void two_shot_ag_kernel(ARParams p, unsigned long long shard_off,
                                   T* __restrict__ out, int64_t numel) {
  __shared__ int s_ok;
  if (threadIdx.x == 0) s_ok = 1;
  __syncthreads();
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    __threadfence_system();
    atomicExch(ar_flag(p.bufs[p.rank]) + 1, p.seq);
  }
  if (threadIdx.x < kMaxRanks && (int)threadIdx.x < p.world) {
    int r = threadIdx.x;
    unsigned long long it = 0;
    while (atomicAdd(ar_flag(p.bufs[r]) + 1, 0ull) < p.seq) {
      if (++it > p.spin_limit) {
        atomicExch(p.error_flag, 1);
        s_ok = 0;
        break;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }
  __syncthreads();
  __threadfence_system();
  if (!s_ok) return;
  int64_t per = (numel + p.world - 1) / p.world;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < numel;
       i += (int64_t)gridDim.x * blockDim.x) {
    int r = (int)(i / per);
    const T* src = (const T*)(p.bufs[r] + shard_off);
    out[i] = src[i - (int64_t)r * per];
  }
}

}  // namespace fi

extern "C" hipError_t fi_two_shot_ar(int dtype, fi::ARParams* p,
                                     unsigned long long shard_off, void* out,
                                     int64_t numel, hipStream_t stream) {
  int grid = (int)((numel / (p->world > 0 ? p->world : 1) + 255) / 256);
  if (grid > 256) grid = 256;
  if (grid == 0) grid = 1;
  dim3 g(grid), g2(grid * (p->world > 0 ? p->world : 1) > 256
                       ? 256
                       : grid * (p->world > 0 ? p->world : 1));
  dim3 blk(256);
#define LTS(T)                                                                   \
  do {                                                                           \
    hipLaunchKernelGGL((fi::two_shot_rs_kernel<T>), g, blk, 0, stream, *p,       \
                       shard_off, numel);                                        \
    hipLaunchKernelGGL((fi::two_shot_ag_kernel<T>), g2, blk, 0, stream, *p,      \
                       shard_off, (T*)out, numel);                               \
  } while (0)
  switch (dtype) {
    case 0: LTS(fi::bf16); break;
    case 1: LTS(fi::fp16); break;
    case 2: LTS(float); break;
    default: return hipErrorInvalidValue;
  }
#undef LTS
  return hipGetLastError();
}
