// Vectorized load/store helpers. CDNA4: hipcc does not auto-vectorize bf16
// scalar loads — always move 8–16 B per lane (short4/short8 reinterpret).
#pragma once
#include "fi/common.hpp"

namespace fi {

// vec_t<T, N>: N elements of T moved as one (up to 16-byte) vector memory op,
// with elementwise f32 conversion accessors. This is the working type for all
// memory-bound elementwise/reduction kernels (norm, rope, activation,
// sampling, paging).
template <typename T, int N>
struct vec_t {
  static_assert(sizeof(T) * 8 >= 8, "unsupported");
  T data[N];

  __device__ __forceinline__ void load(const T* ptr);
  __device__ __forceinline__ void store(T* ptr) const;
  __device__ __forceinline__ float get(int i) const { return to_f32<T>(data[i]); }
  __device__ __forceinline__ void set(int i, float v) { data[i] = from_f32<T>(v); }
  __device__ __forceinline__ void fill(float v) {
#pragma unroll
    for (int i = 0; i < N; ++i) data[i] = from_f32<T>(v);
  }
};

// generic load of exactly sizeof(T)*N bytes
template <typename T, int N>
__device__ __forceinline__ void load_bytes(T* dst, const T* src) {
  constexpr int bytes = sizeof(T) * N;
  if constexpr (bytes == 32) {
    reinterpret_cast<uintx4*>(dst)[0] = reinterpret_cast<const uintx4*>(src)[0];
    reinterpret_cast<uintx4*>(dst)[1] = reinterpret_cast<const uintx4*>(src)[1];
  } else if constexpr (bytes == 16) {
    *reinterpret_cast<uintx4*>(dst) = *reinterpret_cast<const uintx4*>(src);
  } else if constexpr (bytes == 8) {
    *reinterpret_cast<uint64_t*>(dst) = *reinterpret_cast<const uint64_t*>(src);
  } else if constexpr (bytes == 4) {
    *reinterpret_cast<uint32_t*>(dst) = *reinterpret_cast<const uint32_t*>(src);
  } else if constexpr (bytes == 2) {
    *reinterpret_cast<uint16_t*>(dst) = *reinterpret_cast<const uint16_t*>(src);
  } else {
#pragma unroll
    for (int i = 0; i < N; ++i) dst[i] = src[i];
  }
}

template <typename T, int N>
__device__ __forceinline__ void vec_t<T, N>::load(const T* ptr) {
  load_bytes<T, N>(data, ptr);
}
template <typename T, int N>
__device__ __forceinline__ void vec_t<T, N>::store(T* ptr) const {
  load_bytes<T, N>(ptr, data);
}

}  // namespace fi
