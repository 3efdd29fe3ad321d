// Intra-kernel event profiler for gfx950 (reference parity:
// flashinfer/profiler/__init__.py + include/flashinfer/profiler.cuh role).
// Kernels append (tag, timestamp) events into a uint64 ring buffer when the
// caller passes one; a null buffer costs one SGPR branch per event site.
//
// Buffer layout (uint64 slots):
//   [0] capacity in slots (written by the host)
//   [1] write cursor (device atomic)
//   [2..] events: high 32 bits = tag, low 32 = s_memrealtime() low word.
// Tag layout (matches flashinfer/profiler decode_tag:34 in spirit):
//   bits 0-1 event type (0 begin, 1 end, 2 instant)
//   bits 2-11 event index
//   bits 12-31 flat block id
#pragma once
#include <hip/hip_runtime.h>

namespace fi {

enum class ProfType : uint32_t { kBegin = 0, kEnd = 1, kInstant = 2 };

// Call from all threads; only (threadIdx == 0) writes. The realtime counter
// is the constant-rate wall clock (~100 MHz), comparable across CUs.
__device__ __forceinline__ void prof_event(unsigned long long* buf, uint32_t event,
                                           ProfType type) {
  if (buf == nullptr) return;
  if (threadIdx.x + threadIdx.y + threadIdx.z != 0) return;
  unsigned long long cap = buf[0];
  unsigned long long idx = atomicAdd(buf + 1, 1ull) + 2;
  if (idx >= cap) return;
  uint32_t bid = (uint32_t)(blockIdx.x + gridDim.x * (blockIdx.y + gridDim.y * blockIdx.z));
  uint32_t tag = ((uint32_t)type & 3u) | ((event & 0x3FFu) << 2) | ((bid & 0xFFFFFu) << 12);
  unsigned long long t = __builtin_amdgcn_s_memrealtime();
  buf[idx] = ((unsigned long long)tag << 32) | (t & 0xFFFFFFFFull);
}

}  // namespace fi
