#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(int* o) {
  int lane = threadIdx.x;
  int a = 1000 + lane;    // register A: value encodes lane
  int b = 2000 + lane;    // register B
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  o[lane] = r[0];
  o[64 + lane] = r[1];
}

int main() {
  int* d; hipMalloc(&d, 128 * sizeof(int));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d);
  int h[128]; hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  printf("r0 lo (lanes 0,1,31): %d %d %d | r0 hi (32,33,63): %d %d %d\n",
         h[0], h[1], h[31], h[32], h[33], h[63]);
  printf("r1 lo (lanes 0,1,31): %d %d %d | r1 hi (32,33,63): %d %d %d\n",
         h[64], h[65], h[95], h[96], h[97], h[127]);
  return 0;
}
