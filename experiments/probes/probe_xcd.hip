// Which XCD does each block of a 3-D grid land on? (dispatch mapping probe)
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(int* out, int gx, int gy) {
  if (threadIdx.x != 0) return;
  unsigned xcc;
  asm volatile("s_getreg_b32 %0, hwreg(20, 0, 32)" : "=s"(xcc));
  int lin = (blockIdx.z * gy + blockIdx.y) * gx + blockIdx.x;
  out[lin] = (int)(xcc & 0xF);
}

int main() {
  int gx = 16, gy = 6, gz = 64;
  int n = gx * gy * gz;
  int* d;
  (void)hipMalloc(&d, n * 4);
  hipLaunchKernelGGL(probe, dim3(gx, gy, gz), dim3(64), 0, 0, d, gx, gy);
  if (hipDeviceSynchronize() != hipSuccess) { printf("ERR\n"); return 1; }
  int* h = new int[n];
  (void)hipMemcpy(h, d, n * 4, hipMemcpyDeviceToHost);
  printf("first 48 linear (x-fastest) xcd ids: ");
  for (int i = 0; i < 48; ++i) printf("%d", h[i]);
  printf("\nhistogram: ");
  int c[16] = {0};
  for (int i = 0; i < n; ++i) if (h[i] >= 0 && h[i] < 16) c[h[i]]++;
  for (int i = 0; i < 16; ++i) if (c[i]) printf("xcd%d:%d ", i, c[i]);
  // check hypothesis lin %% 8
  int match = 0;
  for (int i = 0; i < n; ++i) if (h[i] == i % 8) match++;
  printf("\nmatch lin%%8: %d/%d\n", match, n);
  return 0;
}
