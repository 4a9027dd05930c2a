// Batched TN GEMM for expert weight gradients (SURVEY.md §2 hot-op #1 bwd):
//   dW[e][i][j] = sum_m A[e][m][i] * B[e][m][j]
// (A = dh_pre [E,M,ffn] or dy [E,M,d]; B = x or h; C = fp32 grad buffer,
// written DIRECTLY — no bmm round trip, no bf16->fp32 convert, no add pass.)
//
// Both operands are m-major in global memory (the reduction dim is the
// leading one), so staging transposes each [64_m][128_i] tile into an
// i-major LDS image [128][64] that the MFMA fragment reads then consume
// exactly like the NT kernel does. The transposed ds_write_b16 scatter pays
// LDS write conflicts; acceptable v1 — the ds_read_b64_tr_b16 path is the
// follow-up (see guide T10).
#include "common.h"

#define TBI 128
#define TBJ 128
#define TBK 64  // reduction (m) tile
#define TN_THREADS 256

// stage one [TBK rows of m][128 cols] tile TRANSPOSED into LDS image
// [128][TBK]. XOR swizzle on the m-chunk index spreads both the scalar
// transpose writes (which vary only in i>>3 within an instruction) and the
// b128 fragment reads (which vary in i) over distinct banks (guide T2/G4):
//   chunk' = (m>>3) ^ (i & 7) ^ ((i >> 3) & 7)
#define TN_OFF(i, m) ((i) * TBK + (((((m) >> 3) ^ ((i) & 7) ^ (((i) >> 3) & 7)) << 3) | ((m) & 7)))

LAH_DEVINL void stage_tile_t(const __bf16* __restrict__ src, __bf16* lds, int col0,
                             int m0, int ld, int tid) {
#pragma unroll
  for (int it = 0; it < (TBK * 128) / (TN_THREADS * 8); ++it) {
    int idx = it * TN_THREADS + tid;  // 16-byte chunk over [64][128]
    int m = idx >> 4;
    int i16 = idx & 15;
    bf16x8 v = *(const bf16x8*)(src + (long)(m0 + m) * ld + col0 + i16 * 8);
#pragma unroll
    for (int q = 0; q < 8; ++q) lds[TN_OFF(i16 * 8 + q, m)] = v[q];
  }
}

template <bool ACC>
__launch_bounds__(TN_THREADS, 2) __global__
void batched_gemm_tn_kernel(const __bf16* __restrict__ A,  // [E, M, I]
                            const __bf16* __restrict__ B,  // [E, M, J]
                            float* __restrict__ C,         // [E, I, J]
                            int M, int I, int J) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[2 * (TBI + TBJ) * TBK];
  const int HALF = (TBI + TBJ) * TBK;

  const int e = blockIdx.z;
  const int j0 = blockIdx.x * TBJ;
  const int i0 = blockIdx.y * TBI;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  const __bf16* Ae = A + (long)e * M * I;
  const __bf16* Be = B + (long)e * M * J;

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const int nk = M / TBK;
  int cur = 0;
  stage_tile_t(Ae, smem, i0, 0, I, tid);
  stage_tile_t(Be, smem + TBI * TBK, j0, 0, J, tid);
  __syncthreads();

  const int l15 = lane & 15;
  const int lk8 = (lane >> 4) * 8;

  for (int t = 0; t < nk; ++t) {
    const __bf16* la = smem + cur * HALF;
    const __bf16* lb = smem + cur * HALF + TBI * TBK;
    if (t + 1 < nk) {
      stage_tile_t(Ae, smem + (cur ^ 1) * HALF, i0, (t + 1) * TBK, I, tid);
      stage_tile_t(Be, smem + (cur ^ 1) * HALF + TBI * TBK, j0, (t + 1) * TBK, J, tid);
    }
#pragma unroll
    for (int kk = 0; kk < TBK / 32; ++kk) {
      bf16x8 afrag[4], bfrag[4];
      const int km = kk * 32 + lk8;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr * 64 + mi * 16 + l15;
        afrag[mi] = *(const bf16x8*)(la + TN_OFF(row, km));
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc * 64 + ni * 16 + l15;
        bfrag[ni] = *(const bf16x8*)(lb + TN_OFF(row, km));
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[mi], bfrag[ni],
                                                                acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // direct fragment-mapped epilogue (an LDS-staged variant was A/B-tested
  // and LOST here: the TN kernel is HBM-bound, and the fp32 LDS roundtrip +
  // 2 barriers cost more than the store coalescing gains)
  float* Ce = C + (long)e * I * J;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = j0 + wc * 64 + ni * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i0 + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
        long off = (long)row * J + col;
        Ce[off] = ACC ? Ce[off] + acc[mi][ni][r] : acc[mi][ni][r];
      }
    }
  }
}

extern "C" void lah_batched_gemm_tn(const void* A, const void* B, void* C, int E, int M,
                                    int I, int J, int accumulate, hipStream_t stream) {
  dim3 grid(J / TBJ, I / TBI, E), block(TN_THREADS);
  if (accumulate)
    hipLaunchKernelGGL(batched_gemm_tn_kernel<true>, grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (float*)C, M, I, J);
  else
    hipLaunchKernelGGL(batched_gemm_tn_kernel<false>, grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (float*)C, M, I, J);
}
