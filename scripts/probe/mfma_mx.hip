// Probe: mfma_scale_f32_32x32x64_f8f6f4 with PER-LANE e8m0 scales.
// Layout hypothesis: lane's A frag = A[row=lane&31][k=(lane>>5)*32 .. +32]
// (one 32-elem MX block) -> one scale byte per lane (opsel 0), value e8m0
// (biased-127 power of two), applied to that lane's block.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
typedef __attribute__((ext_vector_type(8))) int intx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

__global__ void mx_kernel(const unsigned char* A, const unsigned char* B,
                          const unsigned char* sa, const unsigned char* sb,
                          float* C) {
  int lane = threadIdx.x;
  int row = lane & 31, kh = (lane >> 5) * 32;
  intx8 a = *(const intx8*)(A + row * 64 + kh);
  intx8 b = *(const intx8*)(B + row * 64 + kh);
  // block-association probe: FI_SWAP=1 tries lane 32b+m -> block (1-b)
#ifdef SWAP_BLOCK
  int va = sa[row * 2 + 1 - (lane >> 5)];
  int vb = sb[row * 2 + 1 - (lane >> 5)];
#else
  int va = sa[row * 2 + (lane >> 5)];
  int vb = sb[row * 2 + (lane >> 5)];
#endif
  floatx16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, va, 0, vb);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[m * 32 + (lane & 31)] = c[r];
  }
}

static float fp8_to_f32(unsigned char v) {
  int s = v >> 7, e = (v >> 3) & 15, m = v & 7;
  float f;
  if (e == 0) f = ldexpf((float)m / 8.f, -6);
  else if (e == 15 && m == 7) f = NAN;
  else f = ldexpf(1.f + (float)m / 8.f, e - 7);
  return s ? -f : f;
}

int main() {
  unsigned char hA[32 * 64], hB[32 * 64], hsa[64], hsb[64];
  srand(7);
  for (int i = 0; i < 32 * 64; ++i) { hA[i] = rand() & 0x7f; hB[i] = rand() & 0x7f; }
  for (int i = 0; i < 64; ++i) { hsa[i] = 120 + rand() % 16; hsb[i] = 120 + rand() % 16; }
  unsigned char *dA, *dB, *dsa, *dsb; float* dC;
  hipMalloc(&dA, sizeof hA); hipMalloc(&dB, sizeof hB);
  hipMalloc(&dsa, 64); hipMalloc(&dsb, 64); hipMalloc(&dC, 32 * 32 * 4);
  hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
  hipMemcpy(dsa, hsa, 64, hipMemcpyHostToDevice);
  hipMemcpy(dsb, hsb, 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(mx_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dsa, dsb, dC);
  float hC[32 * 32];
  hipMemcpy(hC, dC, sizeof hC, hipMemcpyDeviceToHost);
  // CPU reference: D[m][n] = sum_blk 2^(ea(m,blk)+eb(n,blk)-254) * sum_k A[m][k] B[n][k]
  double maxerr = 0;
  for (int m = 0; m < 32; ++m)
    for (int n = 0; n < 32; ++n) {
      double acc = 0;
      for (int blk = 0; blk < 2; ++blk) {
        double part = 0;
        for (int k = 0; k < 32; ++k)
          part += (double)fp8_to_f32(hA[m * 64 + blk * 32 + k]) *
                  (double)fp8_to_f32(hB[n * 64 + blk * 32 + k]);
        acc += part * ldexp(1.0, (int)hsa[m * 2 + blk] - 127) *
               ldexp(1.0, (int)hsb[n * 2 + blk] - 127);
      }
      double err = fabs(acc - hC[m * 32 + n]) / fmax(1.0, fabs(acc));
      if (err > maxerr) maxerr = err;
    }
  printf("mfma_scale per-lane e8m0 probe: maxrelerr %g %s\n", maxerr,
         maxerr < 1e-5 ? "PASS" : "FAIL");
  return maxerr < 1e-5 ? 0 : 1;
}
