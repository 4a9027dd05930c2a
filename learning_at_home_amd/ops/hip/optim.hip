// Fused asynchronous-SGD step for expert shards (SURVEY.md §2 hot-op #5:
// per-expert optimizer fused into one pass) + batched bf16 transpose that
// maintains the W^T copies the NT-only GEMM plan needs (ffn_gemm.hip).
//
// One pass per expert-stack: read fp32 master + fp32 grad, update master
// (optional momentum / weight decay), write fp32 master and the bf16 compute
// copy. Memory-bound by design: everything is 16-byte vectorized.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f4;

__global__ void sgd_step_kernel(float* __restrict__ master, const float* __restrict__ grad,
                                float* __restrict__ momentum,  // null = plain SGD
                                __bf16* __restrict__ out_bf16, long n, float lr,
                                float mu, float weight_decay) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < n; i += stride) {
    f4 m = *(const f4*)(master + i);
    f4 g = *(const f4*)(grad + i);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      float gq = g[q] + weight_decay * m[q];
      if (momentum != nullptr) {
        float v = momentum[i + q] * mu + gq;
        momentum[i + q] = v;
        gq = v;
      }
      m[q] -= lr * gq;
    }
    *(f4*)(master + i) = m;
    bf16x4 o;
#pragma unroll
    for (int q = 0; q < 4; ++q) o[q] = f2bf(m[q]);
    *(bf16x4*)(out_bf16 + i) = o;
  }
}

// Fused SGD for [E, R, C] matrix params that also writes the TRANSPOSED bf16
// copy in the same pass (saves the transpose kernel's full re-read of the
// bf16 tensor + a launch). Plain SGD + weight decay (momentum goes through
// the 1-D kernel + separate transpose).
__global__ void sgd_step_t_kernel(float* __restrict__ master, const float* __restrict__ grad,
                                  __bf16* __restrict__ out, __bf16* __restrict__ outT,
                                  int R, int C, float lr, float weight_decay) {
  __shared__ __bf16 tile[64][64 + 2];
  int e = blockIdx.z;
  long base = (long)e * R * C;
  int r0 = blockIdx.y * 64, c0 = blockIdx.x * 64;
  int tr = threadIdx.x >> 3;
  int tc = (threadIdx.x & 7) * 8;
#pragma unroll
  for (int s = 0; s < 64; s += 32) {
    int r = tr + s;
    long off = base + (long)(r0 + r) * C + c0 + tc;
    f4 m0 = *(const f4*)(master + off);
    f4 m1 = *(const f4*)(master + off + 4);
    f4 g0 = *(const f4*)(grad + off);
    f4 g1 = *(const f4*)(grad + off + 4);
    bf16x8 o;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      m0[q] -= lr * (g0[q] + weight_decay * m0[q]);
      m1[q] -= lr * (g1[q] + weight_decay * m1[q]);
      o[q] = f2bf(m0[q]);
      o[q + 4] = f2bf(m1[q]);
    }
    *(f4*)(master + off) = m0;
    *(f4*)(master + off + 4) = m1;
    *(bf16x8*)(out + off) = o;
#pragma unroll
    for (int q = 0; q < 8; ++q) tile[tc + q][r] = o[q];
  }
  __syncthreads();
  long baseT = (long)e * R * C;
#pragma unroll
  for (int s = 0; s < 64; s += 32) {
    int r = tr + s;  // row of the transposed tile (original column)
    bf16x8 v;
#pragma unroll
    for (int q = 0; q < 8; ++q) v[q] = tile[r][tc + q];
    *(bf16x8*)(outT + baseT + (long)(c0 + r) * R + r0 + tc) = v;
  }
}

// Master-free SGD: bf16 params updated in place with STOCHASTIC ROUNDING
// (E[rounded] = exact value, so small updates are preserved in expectation
// without an fp32 master copy — 10 B/param of traffic instead of 16) + the
// transposed bf16 copy in the same pass.
// hash_u32 / round_sr moved to common.h (shared with the fused TN+SGD kernel)

__global__ void sgd_step_sr_t_kernel(__bf16* __restrict__ param, const float* __restrict__ grad,
                                     __bf16* __restrict__ paramT, int R, int C, float lr,
                                     float weight_decay, unsigned seed) {
  __shared__ __bf16 tile[64][64 + 2];
  int e = blockIdx.z;
  long base = (long)e * R * C;
  int r0 = blockIdx.y * 64, c0 = blockIdx.x * 64;
  int tr = threadIdx.x >> 3;
  int tc = (threadIdx.x & 7) * 8;
#pragma unroll
  for (int s = 0; s < 64; s += 32) {
    int r = tr + s;
    long off = base + (long)(r0 + r) * C + c0 + tc;
    bf16x8 p = *(const bf16x8*)(param + off);
    f4 g0 = *(const f4*)(grad + off);
    f4 g1 = *(const f4*)(grad + off + 4);
    bf16x8 o;
    unsigned h = hash_u32((unsigned)(off & 0xFFFFFFFFu) ^ seed);
    unsigned h2 = hash_u32(h + 0x9e3779b9u);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      float pv = bf2f(p[q]);
      pv -= lr * (g0[q] + weight_decay * pv);
      o[q] = round_sr(pv, (q < 2 ? h : h2) >> (16 * (q & 1)));
      float pv2 = bf2f(p[q + 4]);
      pv2 -= lr * (g1[q] + weight_decay * pv2);
      o[q + 4] = round_sr(pv2, hash_u32(h + q) >> 8);
    }
    *(bf16x8*)(param + off) = o;
#pragma unroll
    for (int q = 0; q < 8; ++q) tile[tc + q][r] = o[q];
  }
  __syncthreads();
#pragma unroll
  for (int s = 0; s < 64; s += 32) {
    int r = tr + s;
    bf16x8 v;
#pragma unroll
    for (int q = 0; q < 8; ++q) v[q] = tile[r][tc + q];
    *(bf16x8*)(paramT + base + (long)(c0 + r) * R + r0 + tc) = v;
  }
}

// batched transpose: out[e][c][r] = in[e][r][c], bf16, 64x64 LDS tiles
// (+8-element row pad so read-back columns hit distinct banks)
#define TP 64
__global__ void transpose_bf16_kernel(const __bf16* __restrict__ in, __bf16* __restrict__ out,
                                      int R, int C) {
  __shared__ __bf16 tile[TP][TP + 2];
  int e = blockIdx.z;
  int r0 = blockIdx.y * TP, c0 = blockIdx.x * TP;
  const __bf16* ine = in + (long)e * R * C;
  __bf16* oute = out + (long)e * R * C;
  // 256 threads: each loads 16 rows x 8 cols (bf16x8)
  int tr = threadIdx.x >> 3;          // 0..31  (row within tile, x2)
  int tc = (threadIdx.x & 7) * 8;     // col within tile (8-elem chunks)
#pragma unroll
  for (int s = 0; s < TP; s += 32) {
    int r = tr + s;
    bf16x8 v = *(const bf16x8*)(ine + (long)(r0 + r) * C + c0 + tc);
#pragma unroll
    for (int q = 0; q < 8; ++q) tile[tc + q][r] = v[q];
  }
  __syncthreads();
#pragma unroll
  for (int s = 0; s < TP; s += 32) {
    int r = tr + s;  // now a row of the TRANSPOSED tile (original column)
    bf16x8 v;
#pragma unroll
    for (int q = 0; q < 8; ++q) v[q] = tile[r][tc + q];
    *(bf16x8*)(oute + (long)(c0 + r) * R + r0 + tc) = v;
  }
}

// column sum for bias grads: db[e][c] = sum_m d[e][m][c], fp32 out.
// Each thread owns 8 consecutive columns (bf16x8 loads, 16 B/lane) and the
// m-loop is unrolled 4 deep to keep several loads in flight.
__global__ void colsum_kernel(const __bf16* __restrict__ d, float* __restrict__ db,
                              int M, int C) {
  int e = blockIdx.z;
  int c8 = blockIdx.x * blockDim.x + threadIdx.x;  // 8-column chunk index
  if (c8 * 8 >= C) return;
  const __bf16* de = d + (long)e * M * C + c8 * 8;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int m = 0;
  for (; m + 4 <= M; m += 4) {
    bf16x8 v0 = *(const bf16x8*)(de + (long)(m + 0) * C);
    bf16x8 v1 = *(const bf16x8*)(de + (long)(m + 1) * C);
    bf16x8 v2 = *(const bf16x8*)(de + (long)(m + 2) * C);
    bf16x8 v3 = *(const bf16x8*)(de + (long)(m + 3) * C);
#pragma unroll
    for (int q = 0; q < 8; ++q)
      acc[q] += bf2f(v0[q]) + bf2f(v1[q]) + bf2f(v2[q]) + bf2f(v3[q]);
  }
  for (; m < M; ++m) {
    bf16x8 v = *(const bf16x8*)(de + (long)m * C);
#pragma unroll
    for (int q = 0; q < 8; ++q) acc[q] += bf2f(v[q]);
  }
  float* out = db + (long)e * C + c8 * 8;
#pragma unroll
  for (int q = 0; q < 8; q += 4) *(f4*)(out + q) = *(f4*)(acc + q);
}

extern "C" void lah_sgd_step(void* master, const void* grad, void* momentum,
                             void* out_bf16, long n, float lr, float mu,
                             float weight_decay, hipStream_t stream) {
  long blocks = (n / 4 + 255) / 256;
  if (blocks > 2048) blocks = 2048;  // grid-stride (guide G11)
  hipLaunchKernelGGL(sgd_step_kernel, dim3((unsigned)blocks), dim3(256), 0, stream,
                     (float*)master, (const float*)grad, (float*)momentum,
                     (__bf16*)out_bf16, n, lr, mu, weight_decay);
}

extern "C" void lah_sgd_step_t(void* master, const void* grad, void* out, void* outT,
                               int E, int R, int C, float lr, float weight_decay,
                               hipStream_t stream) {
  dim3 grid(C / 64, R / 64, E), block(256);
  hipLaunchKernelGGL(sgd_step_t_kernel, grid, block, 0, stream, (float*)master,
                     (const float*)grad, (__bf16*)out, (__bf16*)outT, R, C, lr,
                     weight_decay);
}

extern "C" void lah_sgd_step_sr_t(void* param, const void* grad, void* paramT, int E,
                                  int R, int C, float lr, float weight_decay,
                                  unsigned seed, hipStream_t stream) {
  dim3 grid(C / 64, R / 64, E), block(256);
  hipLaunchKernelGGL(sgd_step_sr_t_kernel, grid, block, 0, stream, (__bf16*)param,
                     (const float*)grad, (__bf16*)paramT, R, C, lr, weight_decay, seed);
}

extern "C" void lah_transpose_bf16(const void* in, void* out, int E, int R, int C,
                                   hipStream_t stream) {
  dim3 grid(C / TP, R / TP, E), block(256);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, block, 0, stream, (const __bf16*)in,
                     (__bf16*)out, R, C);
}

extern "C" void lah_colsum(const void* d, void* db, int E, int M, int C,
                           hipStream_t stream) {
  dim3 grid(cdiv(C / 8, 256), 1, E), block(256);
  hipLaunchKernelGGL(colsum_kernel, grid, block, 0, stream, (const __bf16*)d, (float*)db,
                     M, C);
}
