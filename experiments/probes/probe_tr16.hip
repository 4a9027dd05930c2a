#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned short* out, int mode) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[1024];
  int tid = threadIdx.x;
  for (int i = tid; i < 1024; i += 64) lds[i] = (unsigned short)(i + 1000);
  __syncthreads();
  unsigned base = (unsigned)(unsigned long long)(&lds[0]);  // real LDS offset
  int addr = base;  // BYTE address
  if (mode == 1) addr = base + tid * 8;
  else if (mode == 2) addr = base + (tid & 15) * 8;
  else if (mode == 3) addr = base + (tid >> 4) * 128;
  else if (mode == 4) addr = base + ((tid & 15) + (tid >> 4) * 64) * 2;
  unsigned long long v = 0;
  if (mode == 9) {  // sanity: plain ds_read_b64 at lane*8
    addr = base + tid * 8;
    asm volatile("ds_read_b64 %0, %1\n\ts_waitcnt lgkmcnt(0)" : "=v"(v) : "v"(addr));
  } else {
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)" : "=v"(v) : "v"(addr));
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = (unsigned short)(v >> (16 * j));
}

int main() {
  unsigned short* d;
  if (hipMalloc(&d, 64 * 4 * 2) != hipSuccess) { printf("malloc fail\n"); return 1; }
  unsigned short h[256];
  int modes[] = {9, 0, 1, 2, 3, 4};
  for (int mi = 0; mi < 6; ++mi) {
    int mode = modes[mi];
    (void)hipMemset(d, 0xEE, 64 * 4 * 2);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipError_t e = hipDeviceSynchronize();
    if (e != hipSuccess) { printf("mode %d: ERROR %s\n", mode, hipGetErrorString(e)); continue; }
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l)
      printf("  lane %2d: %5d %5d %5d %5d\n", l, h[l*4]-1000, h[l*4+1]-1000, h[l*4+2]-1000, h[l*4+3]-1000);
  }
  return 0;
}
