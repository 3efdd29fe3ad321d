// Which output rows does lane L's sa byte affect? One encoded lane per run.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
typedef __attribute__((ext_vector_type(8))) int intx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

__global__ void probe(const unsigned char* A, const unsigned char* B, float* C,
                      int enc_lane) {
  int lane = threadIdx.x;
  int row = lane & 31, kh = (lane >> 5) * 32;
  intx8 a = *(const intx8*)(A + row * 64 + kh);
  intx8 b = *(const intx8*)(B + row * 64 + kh);
  int sa = (lane == enc_lane) ? (int)0x87878787u : 127;  // 0x87 = 135 -> x2^8
  floatx16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa, 0, 127);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[m * 32 + (lane & 31)] = c[r];
  }
}

int main() {
  unsigned char hA[32 * 64], hB[32 * 64];
  for (int i = 0; i < 32 * 64; ++i) { hA[i] = 0x38; hB[i] = 0x38; }
  unsigned char *dA, *dB; float* dC;
  (void)hipMalloc(&dA, sizeof hA); (void)hipMalloc(&dB, sizeof hB);
  (void)hipMalloc(&dC, 4096);
  (void)hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
  int lanes[] = {0, 1, 2, 15, 16, 31, 32, 33, 47, 48, 62, 63};
  for (int t = 0; t < 12; ++t) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dC, lanes[t]);
    float hC[1024];
    (void)hipMemcpy(hC, dC, sizeof hC, hipMemcpyDeviceToHost);
    printf("lane %2d affects rows (log2 boost, col0):", lanes[t]);
    for (int m = 0; m < 32; ++m) {
      double v = hC[m * 32] / 64.0;  // both blocks of ones: base 64
      int b = (int)lround(log2(v));
      if (b != 0) printf(" m%d:+%d", m, b);
    }
    printf("\n");
  }
  return 0;
}
