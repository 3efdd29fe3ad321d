// Branch-free unsigned division by a runtime constant (page_size, group_size,
// head counts). Hacker's Delight magicu scheme; precomputed on host, used in
// device hot loops. Equivalent role to the reference's fastdiv.cuh.
#pragma once
#include <stdint.h>

#ifndef __host__
#define __host__
#define __device__
#define __forceinline__ inline
#endif

namespace fi {

struct uint_fastdiv {
  uint32_t d;  // divisor
  uint32_t m;  // magic multiplier
  uint32_t s;  // shift
  uint32_t a;  // add-indicator (magic overflowed 32 bits)

  __host__ __device__ uint_fastdiv() : d(1), m(0), s(0), a(0) {}

  // Host-side precompute (Hacker's Delight 10-2, magicu).
  explicit uint_fastdiv(uint32_t divisor) : d(divisor), a(0) {
    if (divisor == 1) { m = 0; s = 0; a = 0; return; }
    int p;
    uint32_t nc, delta, q1, r1, q2, r2;
    nc = (uint32_t)(-1) - ((uint32_t)(-(int32_t)divisor)) % divisor;
    p = 31;
    q1 = 0x80000000u / nc;
    r1 = 0x80000000u - q1 * nc;
    q2 = 0x7FFFFFFFu / divisor;
    r2 = 0x7FFFFFFFu - q2 * divisor;
    do {
      p = p + 1;
      if (r1 >= nc - r1) {
        q1 = 2 * q1 + 1;
        r1 = 2 * r1 - nc;
      } else {
        q1 = 2 * q1;
        r1 = 2 * r1;
      }
      if (r2 + 1 >= divisor - r2) {
        if (q2 >= 0x7FFFFFFFu) a = 1;
        q2 = 2 * q2 + 1;
        r2 = 2 * r2 + 1 - divisor;
      } else {
        if (q2 >= 0x80000000u) a = 1;
        q2 = 2 * q2;
        r2 = 2 * r2 + 1;
      }
      delta = divisor - 1 - r2;
    } while (p < 64 && (q1 < delta || (q1 == delta && r1 == 0)));
    m = q2 + 1;
    s = p - 32;
  }

  __host__ __device__ __forceinline__ uint32_t div(uint32_t n) const {
    if (d == 1) return n;
#if defined(__HIP_DEVICE_COMPILE__)
    uint32_t q = __umulhi(n, m);
#else
    uint32_t q = (uint32_t)(((uint64_t)n * m) >> 32);
#endif
    if (a) {
      uint32_t t = ((n - q) >> 1) + q;
      return t >> (s - 1);
    }
    return q >> s;
  }

  __host__ __device__ __forceinline__ void divmod(uint32_t n, uint32_t& q, uint32_t& r) const {
    q = div(n);
    r = n - q * d;
  }
};

}  // namespace fi
