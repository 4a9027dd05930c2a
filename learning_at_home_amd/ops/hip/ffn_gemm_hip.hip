#include "hip/hip_runtime.h"
// Batched expert-FFN GEMM kernels for MI355X (gfx950) — the DMoE hot path
// (SURVEY.md §2 hot-op #1: expert FFN fwd/bwd as MFMA/LDS-tiled HIP kernels).
//
// One kernel family: batched NT GEMM C[e] = A[e] @ B[e]^T with fused
// epilogues, covering all four activation-side GEMMs of the expert FFN:
//   fwd1: h   = gelu(x @ W1^T + b1)          (EPI_BIAS_GELU, stores pre-act)
//   fwd2: y   = h @ W2^T + b2                (EPI_BIAS)
//   bwd1: dhp = (dy @ W2t^T) * gelu'(pre)    (EPI_GELU_GRAD, reads pre-act)
//   bwd2: dx  = dhp @ W1t^T                  (EPI_NONE)
// (W1t/W2t are transposed bf16 weight copies maintained by the fused
// optimizer so every GEMM is the MFMA-natural NT layout — both operands
// K-major; see optim.hip.)
//
// Structure = the "step-3" ladder structure of the CDNA4 guide (§5):
// 128x128 tile, BK=64, 4 waves of 64x64, mfma_f32_16x16x32_bf16,
// double-buffered LDS staged by global_load_lds (16 B/lane), prefetch
// next K-tile before computing current. Weight-tile (B) rows are L2-reused
// across the M-tiles of the same expert.
#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define NTHREADS 256

#define EPI_NONE 0
#define EPI_BIAS 1
#define EPI_BIAS_GELU 2   // C = gelu(acc + bias), PRE = acc + bias
#define EPI_GELU_GRAD 3   // C = acc * gelu'(PRE)

// stage one [rows x BK] bf16 tile (A: rows from m0 with clamp to M; B: rows
// from n0, N must be a multiple of BN) into LDS, lane-linear image [row][BK].
//
// LDS XOR swizzle (guide T2 + rule 21): ds_read_b128 of column slices from a
// row-major [128][64] bf16 image is 8-way bank-conflicted (row stride 128 B
// ≡ 0 mod the 256-B bank row). global_load_lds writes lane-linear, so the
// swizzle lives on the per-lane SOURCE chunk index (c8 ^ (row & 7)) with the
// matching XOR applied at fragment-read time (lds_chunk below) — the two
// permutations are the same involution.
#define LDS_CHUNK(row, c) (((row) * 8 + ((c) ^ ((row) & 7))) * 8)

template <bool CLAMP>
LAH_DEVINL void stage_tile(const __bf16* __restrict__ src, __bf16* lds, int row0,
                           int max_row, int ld, int k0, int tid) {
#pragma unroll
  for (int i = 0; i < (BM * BK) / (NTHREADS * 8); ++i) {
    int idx = i * NTHREADS + tid;       // 16-byte chunk index
    int row = idx >> 3;                 // BK=64 -> 8 chunks per row
    int c8 = (idx & 7) ^ (row & 7);     // inverse swizzle on the source side
    int grow = row0 + row;
    if (CLAMP) grow = min(grow, max_row - 1);
    glds16(src + (long)grow * ld + k0 + c8 * 8, lds + idx * 8);
  }
}

template <int EPI>
__launch_bounds__(NTHREADS, 2) __global__
void batched_gemm_nt_kernel(const __bf16* __restrict__ A,  // [E, M, K]
                            const __bf16* __restrict__ B,  // [E, N, K]
                            __bf16* __restrict__ C,        // [E, M, N]
                            const float* __restrict__ bias,  // [E, N] or null
                            __bf16* __restrict__ PRE,      // [E, M, N] store/read
                            int M, int N, int K) {
  __shared__ __bf16 smem[2 * (BM + BN) * BK];
  const int HALF = (BM + BN) * BK;

  const int e = blockIdx.z;
  const int n0 = blockIdx.x * BN;
  const int m0 = blockIdx.y * BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;  // wave's 64x64 sub-tile

  const __bf16* Ae = A + (long)e * M * K;
  const __bf16* Be = B + (long)e * N * K;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;
  int cur = 0;
  stage_tile<true>(Ae, smem, m0, M, K, 0, tid);
  stage_tile<false>(Be, smem + BM * BK, n0, N, K, 0, tid);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int l15 = lane & 15;
  const int lk8 = (lane >> 4) * 8;

  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) {
      stage_tile<true>(Ae, smem + (cur ^ 1) * HALF, m0, M, K, (t + 1) * BK, tid);
      stage_tile<false>(Be, smem + (cur ^ 1) * HALF + BM * BK, n0, N, K, (t + 1) * BK, tid);
    }
    const __bf16* la = smem + cur * HALF;
    const __bf16* lb = smem + cur * HALF + BM * BK;
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 afrag[4], bfrag[4];
      const int kchunk = (kk * 32 + lk8) >> 3;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr * 64 + mi * 16 + l15;
        afrag[mi] = *(const bf16x8*)(la + LDS_CHUNK(row, kchunk));
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wc * 64 + ni * 16 + l15;
        bfrag[ni] = *(const bf16x8*)(lb + LDS_CHUNK(row, kchunk));
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[mi], bfrag[ni],
                                                                acc[mi][ni], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: C/D fragment mapping for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg  (cdna_hip_programming.md §3)
  __bf16* Ce = C + (long)e * M * N;
  __bf16* Pe = (EPI == EPI_BIAS_GELU || EPI == EPI_GELU_GRAD)
                   ? PRE + (long)e * M * N : nullptr;
  const float* be = (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) ? bias + (long)e * N : nullptr;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = n0 + wc * 64 + ni * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        long off = (long)row * N + col;
        float v = acc[mi][ni][r];
        if (EPI == EPI_BIAS) {
          Ce[off] = f2bf(v + be[col]);
        } else if (EPI == EPI_BIAS_GELU) {
          float pre = v + be[col];
          Pe[off] = f2bf(pre);
          Ce[off] = f2bf(gelu_fwd(pre));
        } else if (EPI == EPI_GELU_GRAD) {
          Ce[off] = f2bf(v * gelu_bwd(bf2f(Pe[off])));
        } else {
          Ce[off] = f2bf(v);
        }
      }
    }
  }
}

extern "C" void lah_batched_gemm_nt(const void* A, const void* B, void* C,
                                    const void* bias, void* PRE, int E, int M, int N,
                                    int K, int epi, hipStream_t stream) {
  dim3 grid(N / BN, cdiv(M, BM), E);
  dim3 block(NTHREADS);
  switch (epi) {
    case EPI_NONE:
      hipLaunchKernelGGL(batched_gemm_nt_kernel<EPI_NONE>, grid, block, 0, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                         (const float*)bias, (__bf16*)PRE, M, N, K);
      break;
    case EPI_BIAS:
      hipLaunchKernelGGL(batched_gemm_nt_kernel<EPI_BIAS>, grid, block, 0, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                         (const float*)bias, (__bf16*)PRE, M, N, K);
      break;
    case EPI_BIAS_GELU:
      hipLaunchKernelGGL(batched_gemm_nt_kernel<EPI_BIAS_GELU>, grid, block, 0, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                         (const float*)bias, (__bf16*)PRE, M, N, K);
      break;
    case EPI_GELU_GRAD:
      hipLaunchKernelGGL(batched_gemm_nt_kernel<EPI_GELU_GRAD>, grid, block, 0, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                         (const float*)bias, (__bf16*)PRE, M, N, K);
      break;
    default:
      abort();
  }
}
