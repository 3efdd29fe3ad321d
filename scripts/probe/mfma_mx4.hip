#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
typedef __attribute__((ext_vector_type(8))) int intx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

template <int OPSEL_A>
__global__ void probe(const unsigned char* A, const unsigned char* B,
                      float* C, int which) {
  int lane = threadIdx.x;
  int row = lane & 31, kh = (lane >> 5) * 32;
  intx8 a = *(const intx8*)(A + row * 64 + kh);
  intx8 b = *(const intx8*)(B + row * 64 + kh);
  // bytes encode their own index: byte b = 100 + 8*b (same on all lanes)
  unsigned int enc = 100u | (108u << 8) | (116u << 16) | (124u << 24);
  int sa = which == 0 ? (int)enc : 127;
  int sb = which == 1 ? (int)enc : 127;
  floatx16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, OPSEL_A, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[m * 32 + (lane & 31)] = c[r];
  }
}

int main() {
  unsigned char hA[32 * 64], hB[32 * 64];
  for (int blkcase = 0; blkcase < 2; ++blkcase) {
    for (int i = 0; i < 32 * 64; ++i) {
      int k = i % 64;
      hA[i] = ((k / 32) == blkcase) ? 0x38 : 0;
      hB[i] = 0x38;
    }
    unsigned char *dA, *dB; float* dC;
    (void)hipMalloc(&dA, sizeof hA); (void)hipMalloc(&dB, sizeof hB);
    (void)hipMalloc(&dC, 32 * 32 * 4);
    (void)hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
    (void)hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
    for (int which = 0; which < 2; ++which) {
      hipLaunchKernelGGL(probe<0>, dim3(1), dim3(64), 0, 0, dA, dB, dC, which);
      // also opsel 1..3 for which==0

      float hC[32 * 32];
      (void)hipMemcpy(hC, dC, sizeof hC, hipMemcpyDeviceToHost);
      printf("Ablk %d enc %s: m0=%d m1=%d m16=%d (value-100)/8 = byte\n",
             blkcase, which == 0 ? "A" : "B",
             (int)lround(log2(hC[0] / 32.0)) + 27,
             (int)lround(log2(hC[32] / 32.0)) + 27,
             (int)lround(log2(hC[16 * 32] / 32.0)) + 27);
    }
    (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dC);
  }
  return 0;
}
