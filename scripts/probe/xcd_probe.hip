// XCD placement probe: records each workgroup's XCC_ID for a
// decode-split-shaped grid (x=batch, y=kv_heads, z=split) to test whether
// the dispatcher round-robins consecutive linear workgroup ids across the 8
// XCDs — the precondition for the same-XCD in-kernel split merge.
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(int* out) {
  if (threadIdx.x == 0) {
    uint32_t xcc;
    asm volatile("s_getreg_b32 %0, hwreg(HW_REG_XCC_ID)" : "=s"(xcc));
    int id = blockIdx.x + gridDim.x * (blockIdx.y + gridDim.y * blockIdx.z);
    out[id] = (int)(xcc & 0xf);
  }
}

int main() {
  for (int bx : {16, 8, 12}) {
    dim3 g(bx, 8, 4);
    int n = g.x * g.y * g.z;
    int* d;
    hipMalloc(&d, n * sizeof(int));
    hipMemset(d, 0xff, n * sizeof(int));
    hipLaunchKernelGGL(probe, g, dim3(512), 0, 0, d);
    hipDeviceSynchronize();
    int* h = new int[n];
    hipMemcpy(h, d, n * sizeof(int), hipMemcpyDeviceToHost);
    // check: for each (x,y), do all z share an XCD?
    int bad = 0, mod8_ok = 0;
    for (int i = 0; i < n; ++i)
      if (h[i] == i % 8) mod8_ok++;
    for (int y = 0; y < 8; ++y)
      for (int x = 0; x < bx; ++x) {
        int x0 = h[x + bx * y];
        for (int z = 1; z < 4; ++z)
          if (h[x + bx * (y + 8 * z)] != x0) bad++;
      }
    printf("grid(%d,8,4): linear-id mod 8 == XCC for %d/%d; "
           "(x,y) groups with mismatched z: %d\n", bx, mod8_ok, n, bad);
    printf("  first 24 ids' xcc: ");
    for (int i = 0; i < 24; ++i) printf("%d", h[i]);
    printf("\n");
    hipFree(d);
    delete[] h;
  }
  return 0;
}
