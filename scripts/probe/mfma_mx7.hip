#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
typedef __attribute__((ext_vector_type(8))) int intx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

__global__ void probe(const unsigned char* A, const unsigned char* B,
                      const unsigned char* sa_tab, float* C) {
  int lane = threadIdx.x;
  int row = lane & 31, kh = (lane >> 5) * 32;
  intx8 a = *(const intx8*)(A + row * 64 + kh);
  intx8 b = *(const intx8*)(B + row * 64 + kh);
  int va = sa_tab[row * 2 + (lane >> 5)];  // natural (row, block) layout
  floatx16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, va, 0, 127);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[m * 32 + (lane & 31)] = c[r];
  }
}

int main() {
  unsigned char hA[32 * 64], hB[32 * 64], hsa[64];
  srand(3);
  for (int i = 0; i < 64; ++i) hsa[i] = 119 + rand() % 16;
  unsigned char *dA, *dB, *dsa; float* dC;
  (void)hipMalloc(&dA, sizeof hA); (void)hipMalloc(&dB, sizeof hB);
  (void)hipMalloc(&dsa, 64); (void)hipMalloc(&dC, 4096);
  (void)hipMemcpy(dsa, hsa, 64, hipMemcpyHostToDevice);
  for (int i = 0; i < 32 * 64; ++i) hB[i] = 0x38;
  (void)hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
  for (int blk = 0; blk < 2; ++blk) {
    for (int i = 0; i < 32 * 64; ++i) hA[i] = ((i % 64) / 32 == blk) ? 0x38 : 0;
    (void)hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dsa, dC);
    float hC[1024];
    (void)hipMemcpy(hC, dC, sizeof hC, hipMemcpyDeviceToHost);
    printf("blk%d: m(applied | sa[m][0] sa[m][1] both-exp):\n", blk);
    for (int m = 0; m < 8; ++m) {
      int applied = (int)lround(log2(hC[m * 32] / 32.0)) + 127;
      printf(" m%d(%d|%d,%d)", m, applied, hsa[2 * m], hsa[2 * m + 1]);
    }
    printf("\n");
  }
  return 0;
}
