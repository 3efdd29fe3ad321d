// Diagnostic: which (lane, byte) does mfma_scale read the A/B e8m0 scale
// from, per (row, k-block)? Encode the lane id in the scale exponent and
// read it back from the output magnitude.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
typedef __attribute__((ext_vector_type(8))) int intx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

__global__ void probe(const unsigned char* A, const unsigned char* B,
                      float* C, int encode_byte) {
  int lane = threadIdx.x;
  int row = lane & 31, kh = (lane >> 5) * 32;
  intx8 a = *(const intx8*)(A + row * 64 + kh);
  intx8 b = *(const intx8*)(B + row * 64 + kh);
  // scale reg: byte `encode_byte` = 100 + lane, others 0x7F (unit)
  unsigned int enc = 0x7F7F7F7Fu;
  enc &= ~(0xFFu << (8 * encode_byte));
  enc |= (unsigned)(100 + lane) << (8 * encode_byte);
  floatx16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, (int)enc,
                                                      0, 127);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[m * 32 + (lane & 31)] = c[r];
  }
}

int main() {
  unsigned char hA[32 * 64], hB[32 * 64];
  // fp8 1.0 = 0x38
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
    for (int byte = 0; byte < 2; ++byte) {
      hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dC, byte);
      float hC[32 * 32];
      (void)hipMemcpy(hC, dC, sizeof hC, hipMemcpyDeviceToHost);
      printf("A-block %d, enc byte %d: rows 0..7 -> src lane: ", blkcase, byte);
      for (int m = 0; m < 8; ++m) {
        double v = hC[m * 32 + 0] / 32.0;  // col 0
        int srclane = v > 0 ? (int)lround(log2(v)) + 27 : -1;
        printf("%d ", srclane);
      }
      printf("\n");
    }
    (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dC);
  }
  return 0;
}
