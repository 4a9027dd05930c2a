// Batched expert-FFN GEMM kernels for MI355X (gfx950) — the DMoE hot path
// (SURVEY.md §2 hot-op #1: expert FFN fwd/bwd as MFMA/LDS-tiled HIP kernels).
//
// One kernel family: batched NT GEMM C[e] = A[e] @ B[e]^T with fused
// epilogues, covering all four activation-side GEMMs of the expert FFN:
//   fwd1: h   = gelu(x @ W1^T + b1)        (EPI_BIAS_GELU, stores gelu'(pre))
//   fwd2: y   = h @ W2^T + b2              (EPI_BIAS)
//   bwd1: dhp = (dy @ W2t^T) * DER         (EPI_GELU_GRAD, reads gelu'(pre))
//   bwd2: dx  = dhp @ W1t^T                (EPI_NONE)
// The fwd GELU epilogue stores the DERIVATIVE gelu'(pre) (not the pre-act):
// it reuses the tanh it already computed, making the backward epilogue a
// pure multiply — no transcendentals in bwd.
//
// Structure: guide §5 "step-3" + T2 swizzle: LDS-staged via global_load_lds
// (16 B/lane), double-buffered, mfma_f32_16x16x32_bf16, XOR chunk swizzle on
// the glds SOURCE address with the matching XOR at fragment read (rule 21).
// Tile geometry is templated: 128x128 (4 waves of 64x64) for M%128==0,
// 64x128 (4 waves of 64x32) otherwise — capacity buffers are padded to 64
// rows, so a 192-row expert batch runs as 3 full 64-tiles with zero waste.
#include "common.h"
#include <cstdlib>

#define BK 64
#define NTHREADS 256

#define EPI_NONE 0
#define EPI_BIAS 1
#define EPI_BIAS_GELU 2   // C = gelu(acc + bias), DER = gelu'(acc + bias)
#define EPI_GELU_GRAD 3   // C = acc * DER

// swizzled LDS element offset of 16-byte chunk c in row `row` (guide T2)
#define LDS_CHUNK(row, c) (((row) * 8 + ((c) ^ ((row) & 7))) * 8)

template <int ROWS, bool CLAMP, int THREADS = NTHREADS>
LAH_DEVINL void stage_tile(const __bf16* __restrict__ src, __bf16* lds, int row0,
                           int max_row, int ld, int k0, int tid) {
#pragma unroll
  for (int i = 0; i < (ROWS * BK) / (THREADS * 8); ++i) {
    int idx = i * THREADS + tid;        // 16-byte chunk index
    int row = idx >> 3;                 // BK=64 -> 8 chunks per row
    int c8 = (idx & 7) ^ (row & 7);     // inverse swizzle on the source side
    int grow = row0 + row;
    if (CLAMP) grow = min(grow, max_row - 1);
    glds16(src + (long)grow * ld + k0 + c8 * 8, lds + idx * 8);
  }
}

// gelu and its derivative from one tanh evaluation. tanh via raw v_exp_f32:
// t = 1 - 2/(e^{2u}+1). libm tanhf's range/denorm fixup is ~40 cycles per
// call and dominated the EPI_BIAS_GELU epilogue (PMC: 31.5% ACTIVE_INST on
// fwd1 vs 16% on the other variants); the exp2 form is 1-2 f32 ulp off
// libm in the gelu-relevant range, invisible at bf16 output precision.
// Saturation: u -> +inf gives e=inf -> t=1; u -> -inf gives e=0 -> t=-1.
LAH_DEVINL void gelu_both(float x, float* g, float* dg) {
  float x2 = x * x;
  float u = GELU_C0 * (x + GELU_C1 * x * x2);
  float e = __builtin_amdgcn_exp2f(u * 2.885390081777927f);  // 2*log2(e)
  float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(e + 1.0f);
  float du = GELU_C0 * (1.0f + 3.0f * GELU_C1 * x2);
  *g = 0.5f * x * (1.0f + t);
  *dg = 0.5f * (1.0f + t) + 0.5f * x * (1.0f - t * t) * du;
}

// padded fp32-tile column map: a 4-float gap every 16 columns makes both
// the b128 read groups (20k+4q mod 64 distinct per lane, CSTR ≡ 0 mod 64)
// and the b32 scatter writes (<=2-way, free) conflict-free — the unpadded
// layout was a 4-way read conflict worth 6.6e8 SQ_LDS_BANK_CONFLICT on the
// fwd1 epilogue (PMC, flagship shapes)
#define CT_COL(c) ((c) + ((((c) >> 4)) << 2))

// coalesced bf16x8 store pass: ERB rows of the fp32 LDS tile -> global
template <int EPI, int BN, int CSTR, int THREADS>
LAH_DEVINL void emit_rows(const float* ctile, __bf16* Ce, __bf16* De, const float* be,
                          int row0, int n0, int M, int N, int nrows, int tid) {
  constexpr int TPR = BN / 16;  // threads per row (16 cols x bf16x8 pair each)
  const int ecol = (tid % TPR) * 16;
  const int lcol = CT_COL(ecol);
  // bias is row-invariant: load this thread's 16-wide chunk ONCE
  float bv[16];
  if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) {
#pragma unroll
    for (int q = 0; q < 16; q += 4)
      *(f32x4*)(bv + q) = *(const f32x4*)(be + n0 + ecol + q);
  }
  for (int rr = tid / TPR; rr < nrows; rr += THREADS / TPR) {
    int row = row0 + rr;
    if (row >= M) continue;
    long goff = (long)row * N + n0 + ecol;
    // vectorized conflict-free tile reads: 4 x f32x4 per 16-wide chunk
    float cv[16];
#pragma unroll
    for (int q4 = 0; q4 < 4; ++q4)
      *(f32x4*)(cv + q4 * 4) = *(const f32x4*)(ctile + rr * CSTR + lcol + q4 * 4);
    bf16x8 outv[2], derv[2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      bf16x8 dv{};
      if (EPI == EPI_GELU_GRAD) dv = *(const bf16x8*)(De + goff + h * 8);
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        float v = cv[h * 8 + q];
        if (EPI == EPI_BIAS) {
          outv[h][q] = f2bf(v + bv[h * 8 + q]);
        } else if (EPI == EPI_BIAS_GELU) {
          float g, dg;
          gelu_both(v + bv[h * 8 + q], &g, &dg);
          outv[h][q] = f2bf(g);
          derv[h][q] = f2bf(dg);
        } else if (EPI == EPI_GELU_GRAD) {
          outv[h][q] = f2bf(v * bf2f(dv[q]));
        } else {
          outv[h][q] = f2bf(v);
        }
      }
      *(bf16x8*)(Ce + goff + h * 8) = outv[h];
      if (EPI == EPI_BIAS_GELU) *(bf16x8*)(De + goff + h * 8) = derv[h];
    }
  }
}

// WM x WN waves, each computing FM x FN fragments of 16x16:
// BM = WM*FM*16, BN = WN*FN*16. Instantiated as <EPI, 2,2,4,4> (128x128)
// and <EPI, 1,4,4,2> (64x128).
template <int EPI, int WM, int WN, int FM, int FN>
__launch_bounds__(WM * WN * 64, (WM * WN > 4) ? 1 : 2) __global__
void batched_gemm_nt_kernel(const __bf16* __restrict__ A,  // [E, M, K]
                            const __bf16* __restrict__ B,  // [E, N, K]
                            __bf16* __restrict__ C,        // [E, M, N]
                            const float* __restrict__ bias,  // [E, N] or null
                            __bf16* __restrict__ DER,      // [E, M, N] store/read
                            int M, int N, int K, int swz) {
  constexpr int BM = WM * FM * 16;
  constexpr int BN = WN * FN * 16;
  constexpr int THREADS = WM * WN * 64;
  __shared__ __attribute__((aligned(16))) __bf16 smem[2 * (BM + BN) * BK];
  const int HALF = (BM + BN) * BK;

  // Optional fine-grained XCD remap (LAH_XCD_SWZ=1): within each group of
  // 64 hardware blocks (dispatch = lin%8 -> XCD, probe-verified), XCD x
  // takes 8 CONSECUTIVE tiles — neighbor tiles share one L2 while staying
  // temporally co-resident. (A coarse per-XCD mega-chunk remap regressed.)
  int lin = (blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x + blockIdx.x;
  if (swz) {
    int total = gridDim.x * gridDim.y * gridDim.z;
    if (lin < (total & ~63))
      lin = (lin & ~63) | ((lin & 7) << 3) | ((lin >> 3) & 7);
  }
  const int e = lin / (gridDim.x * gridDim.y);
  const int n0 = (lin % gridDim.x) * BN;
  const int m0 = ((lin / gridDim.x) % gridDim.y) * BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave / WN, wc = wave % WN;

  const __bf16* Ae = A + (long)e * M * K;
  const __bf16* Be = B + (long)e * N * K;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;
  int cur = 0;
  stage_tile<BM, true, THREADS>(Ae, smem, m0, M, K, 0, tid);
  stage_tile<BN, false, THREADS>(Be, smem + BM * BK, n0, N, K, 0, tid);

  const int l15 = lane & 15;
  const int lk8 = (lane >> 4) * 8;
  // glds instructions per thread per stage — the counted-vmcnt quantum
  constexpr int GLDS_PER_STAGE = (BM * BK) / (THREADS * 8) + (BN * BK) / (THREADS * 8);

  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) {
      stage_tile<BM, true, THREADS>(Ae, smem + (cur ^ 1) * HALF, m0, M, K, (t + 1) * BK, tid);
      stage_tile<BN, false, THREADS>(Be, smem + (cur ^ 1) * HALF + BM * BK, n0, N, K,
                            (t + 1) * BK, tid);
      // counted wait: stage t landed, stage t+1 stays IN FLIGHT across the
      // barrier (raw s_barrier — __syncthreads would emit vmcnt(0) and drain
      // the prefetch; guide §5 "pipelining across barriers" / T3+T4)
      if constexpr (GLDS_PER_STAGE == 8)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    const __bf16* la = smem + cur * HALF;
    const __bf16* lb = smem + cur * HALF + BM * BK;
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 afrag[FM], bfrag[FN];
      const int kchunk = (kk * 32 + lk8) >> 3;
#pragma unroll
      for (int mi = 0; mi < FM; ++mi) {
        int row = wr * FM * 16 + mi * 16 + l15;
        afrag[mi] = *(const bf16x8*)(la + LDS_CHUNK(row, kchunk));
      }
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
        int row = wc * FN * 16 + ni * 16 + l15;
        bfrag[ni] = *(const bf16x8*)(lb + LDS_CHUNK(row, kchunk));
      }
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[mi], bfrag[ni],
                                                                acc[mi][ni], 0, 0, 0);
    }
    asm volatile("" ::: "memory");
    __builtin_amdgcn_s_barrier();  // all waves done READING buf[cur] before
    cur ^= 1;                      // the next iteration's glds overwrite it
  }

  // epilogue through LDS: scatter the C/D fragments (col = lane&15,
  // row = (lane>>4)*4 + reg — guide §3) into an fp32 LDS image, then emit
  // COALESCED bf16x8 stores (the direct scalar-store epilogue costs ~2 µs
  // per workgroup, ~40% of a K=1024 tile's time). The fp32 tile reuses the
  // staging LDS (BM*BN*4 <= 2*(BM+BN)*BK*2 for both instantiated shapes).
  // rows per epilogue pass: limited by the fp32 LDS image (multi-pass for
  // the 256x256 tile whose full image would need 256 KiB)
  // fp32 LDS image with a +4-float row pad: an unpadded BN-stride put every
  // store-pass lane group on 2 banks (16-way; PMC: 100M SQ_LDS_BANK_CONFLICT
  // on the fwd1 epilogue); stride BN+4 spreads rows by one bank.
  constexpr int CSTR = BN + (BN >> 4) * 4;  // CT_COL-padded stride
  constexpr int LDS_BYTES = 2 * (BM + BN) * BK * 2;
  constexpr int ERB0 = LDS_BYTES / (CSTR * 4);
  constexpr int ERB = (ERB0 >= BM) ? BM : (ERB0 >= 128 ? 128 : (ERB0 >= 64 ? 64 : 32));
  static_assert(ERB >= 16 && BM % ERB == 0, "bad epilogue row split");
  float* ctile = (float*)smem;
#pragma unroll
  for (int ep = 0; ep < BM / ERB; ++ep) {
    __syncthreads();  // staging/previous-pass reads done; smem is free
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
        int col = wc * FN * 16 + ni * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = wr * FM * 16 + mi * 16 + (lane >> 4) * 4 + r;
          if (row / ERB == ep) ctile[(row % ERB) * CSTR + CT_COL(col)] = acc[mi][ni][r];
        }
      }
    }
    __syncthreads();

    __bf16* Ce = C + (long)e * M * N;
    __bf16* De = (EPI == EPI_BIAS_GELU || EPI == EPI_GELU_GRAD)
                     ? DER + (long)e * M * N : nullptr;
    const float* be = (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) ? bias + (long)e * N : nullptr;
    emit_rows<EPI, BN, CSTR, THREADS>(ctile, Ce, De, be, m0 + ep * ERB, n0, M, N, ERB, tid);
  }
}


static int lah_xcd_swz() {
  // default ON since round 2: +1.9% on the flagship step (in-step L2
  // locality; the isolated NT shapes move less). LAH_XCD_SWZ=0 disables.
  static int v = -1;
  if (v < 0) {
    const char* env = getenv("LAH_XCD_SWZ");
    v = (env != nullptr && env[0] == '0') ? 0 : 1;
  }
  return v;
}

// ---------------------------------------------------------------------------
// 8-phase 256x256 schedule (guide §5 template, reconstructed; env-gated
// LAH_GEMM_8P=1, OFF by default). MEASURED: numerically correct (3x race
// screens x2 variants) but 940 TF (reads-after-barrier) / 882 TF (guide
// order: reads+stage pre-barrier, end-of-phase certification waits) vs
// 1064 TF for the 1-phase counted-vmcnt kernel at 4k^3 — the guide's own
// caveat reproduced both ways: a phase split without the exact per-phase
// ds_read || global_load || MFMA instruction interleave HURTS (-7..-27%).
// Kept as the starting point for the fine-interleave follow-up:
// 4 phases per K-tile, one output QUADRANT per phase per wave (16 MFMA),
// one HALF-tile staged per phase (2 glds/thread), counted vmcnt so 2+ halves
// stay in flight across RAW barriers. "Halves" are row-bands grouped by
// consumption: A-half h = rows {h*64+wr*128}, B-half h = cols {h*32+wc*64},
// so phase p=(rh,ch) consumes exactly A-half rh and B-half ch and each slot
// frees mid-tile. Issue order per tile: [Ah0, Bh0, Ah1, Bh1] (of tile t+1);
// waits: vmcnt(4) before phases 0 and 2 (the two oldest halves landed).
// ---------------------------------------------------------------------------
template <int EPI>
__launch_bounds__(512, 1) __global__
void batched_gemm_nt_8p_kernel(const __bf16* __restrict__ A,  // [E, M, K]
                               const __bf16* __restrict__ B,  // [E, N, K]
                               __bf16* __restrict__ C,        // [E, M, N]
                               const float* __restrict__ bias,
                               __bf16* __restrict__ DER,
                               int M, int N, int K) {
  constexpr int BM = 256, BN = 256, THREADS = 512;
  constexpr int WM = 2, WN = 4, FM = 8, FN = 4;
  __shared__ __attribute__((aligned(16))) __bf16 smem[2 * (BM + BN) * BK];
  const int HALF = (BM + BN) * BK;

  const int e = blockIdx.z;
  const int n0 = blockIdx.x * BN;
  const int m0 = blockIdx.y * BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave / WN, wc = wave % WN;

  const __bf16* Ae = A + (long)e * M * K;
  const __bf16* Be = B + (long)e * N * K;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int l15 = lane & 15;
  const int lk8 = (lane >> 4) * 8;
  const int nk = K / BK;

  // stage one A-half (rows {h*64 + band*128}, 128 rows) of K-tile t
  auto stage_a_half = [&](int buf, int t, int h) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * THREADS + tid;           // 16B chunk in the half
      int rl = idx >> 3;                     // 0..127
      int row = (rl & 63) + ((rl >> 6) << 7) + h * 64;  // actual tile row
      int c8 = (idx & 7) ^ (row & 7);
      glds16(Ae + (long)min(m0 + row, M - 1) * K + t * BK + c8 * 8,
             smem + buf * HALF + row * BK + (idx & 7) * 8);
    }
  };
  // stage one B-half (cols {h*32 + wc*64}, 4 bands of 32 rows)
  auto stage_b_half = [&](int buf, int t, int h) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * THREADS + tid;
      int rl = idx >> 3;                     // 0..127
      int row = (rl & 31) + ((rl >> 5) << 6) + h * 32;
      int c8 = (idx & 7) ^ (row & 7);
      glds16(Be + (long)(n0 + row) * K + t * BK + c8 * 8,
             smem + buf * HALF + BM * BK + row * BK + (idx & 7) * 8);
    }
  };

  // prologue: tile 0's halves in issue order; certify Ah0+Bh0 then barrier
  stage_a_half(0, 0, 0);
  stage_b_half(0, 0, 0);
  stage_b_half(0, 0, 1);
  stage_a_half(0, 0, 1);
  if (nk > 1) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  bf16x8 afrag[FM], bfrag[FN];  // current quadrant row/col-half fragments

  for (int t = 0; t < nk; ++t) {
    const __bf16* la = smem + (t & 1) * HALF;
    const __bf16* lb = smem + (t & 1) * HALF + BM * BK;
    const int nb = (t + 1) & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // GRAY quadrant order (0,0),(0,1),(1,1),(1,0): each consumed half was
      // issued >= 3 phases earlier under issue order [Ah0, Bh0, Bh1, Ah1]
      const int rh = p >> 1, ch = (p & 1) ^ (p >> 1);
      if (t + 1 < nk) {
        if (p == 0) stage_a_half(nb, t + 1, 0);
        else if (p == 1) stage_b_half(nb, t + 1, 0);
        else if (p == 2) stage_b_half(nb, t + 1, 1);
        else stage_a_half(nb, t + 1, 1);
      }
      // (the wait certifying THIS phase's halves sat before the PREVIOUS
      // phase's closing barrier — see the phase tail below — so the
      // pre-barrier ds_reads here are safe across waves)
      // guide phase order: ds_reads issued BEFORE the mid barrier (latency
      // drains during barrier arrival skew), lgkmcnt(0) after it
      // quadrant fragments: rows wr*128 + rh*64 (+mi*16), cols wc*64 + ch*32
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int kchunk = (kk * 32 + lk8) >> 3;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          int row = wr * 128 + rh * 64 + mi * 16 + l15;
          afrag[kk * 4 + mi] = *(const bf16x8*)(la + LDS_CHUNK(row, kchunk));
        }
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          int row = wc * 64 + ch * 32 + ni * 16 + l15;
          bfrag[kk * 2 + ni] = *(const bf16x8*)(lb + LDS_CHUNK(row, kchunk));
        }
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[rh * 4 + mi][ch * 2 + ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[kk * 4 + mi], bfrag[kk * 2 + ni], acc[rh * 4 + mi][ch * 2 + ni],
                0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // certify the NEXT phase's halves before the closing barrier: with the
      // per-tile issue order [Ah0,Bh0,Bh1,Ah1], the half first needed by
      // phase p+1 is always "all but the 2 newest in flight" => vmcnt(4)
      // (ends of p0, p1 and p3; nothing new needed entering p3). Tail tiles
      // have no fresh issues covering the count: drain fully.
      if (p != 2) {
        if (t + 1 < nk)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue identical to the main kernel (fp32 LDS image, padded)
  constexpr int CSTR = BN + (BN >> 4) * 4;  // CT_COL-padded stride
  constexpr int ERB = 64;
  float* ctile = (float*)smem;
#pragma unroll
  for (int ep = 0; ep < BM / ERB; ++ep) {
    __syncthreads();
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
        int col = wc * FN * 16 + ni * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = wr * FM * 16 + mi * 16 + (lane >> 4) * 4 + r;
          if (row / ERB == ep) ctile[(row % ERB) * CSTR + CT_COL(col)] = acc[mi][ni][r];
        }
      }
    }
    __syncthreads();
    __bf16* Ce = C + (long)e * M * N;
    __bf16* De = (EPI == EPI_BIAS_GELU || EPI == EPI_GELU_GRAD)
                     ? DER + (long)e * M * N : nullptr;
    const float* be = (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) ? bias + (long)e * N : nullptr;
    emit_rows<EPI, BN, CSTR, THREADS>(ctile, Ce, De, be, m0 + ep * ERB, n0, M, N, ERB, tid);
  }
}

static int lah_gemm_8p() {
  static int v = -1;
  if (v < 0) {
    const char* env = getenv("LAH_GEMM_8P");
    v = (env != nullptr && env[0] == '1') ? 1 : 0;
  }
  return v;
}

// runtime switch for the big-tile variant (A/B tuning; see bench_kernels.py)
static int lah_use_256_tile() {
  static int v = -1;
  if (v < 0) {
    const char* env = getenv("LAH_GEMM_256");
    v = (env == nullptr || env[0] != '0') ? 1 : 0;
  }
  return v;
}

template <int EPI>
static void launch_nt(const void* A, const void* B, void* C, const void* bias,
                      void* DER, int E, int M, int N, int K, hipStream_t stream) {
  if (M % 256 == 0 && N % 256 == 0 && lah_gemm_8p()) {
    dim3 grid(N / 256, M / 256, E), block(512);
    hipLaunchKernelGGL((batched_gemm_nt_8p_kernel<EPI>), grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                       (const float*)bias, (__bf16*)DER, M, N, K);
  } else if (M % 256 == 0 && N % 256 == 0 && lah_use_256_tile()) {
    // 256x256 tile, 8 waves, 128 KiB LDS, 1 block/CU (guide: high-tile regime)
    dim3 grid(N / 256, M / 256, E), block(512);
    hipLaunchKernelGGL((batched_gemm_nt_kernel<EPI, 2, 4, 8, 4>), grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                       (const float*)bias, (__bf16*)DER, M, N, K, lah_xcd_swz());
  } else if (M % 128 == 0) {
    dim3 grid(N / 128, M / 128, E), block(256);
    hipLaunchKernelGGL((batched_gemm_nt_kernel<EPI, 2, 2, 4, 4>), grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                       (const float*)bias, (__bf16*)DER, M, N, K, lah_xcd_swz());
  } else {  // M padded to 64 (capacity buffers): 64-row tiles, zero tail waste
    dim3 grid(N / 128, cdiv(M, 64), E), block(256);
    hipLaunchKernelGGL((batched_gemm_nt_kernel<EPI, 1, 4, 4, 2>), grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                       (const float*)bias, (__bf16*)DER, M, N, K, lah_xcd_swz());
  }
}

extern "C" void lah_batched_gemm_nt(const void* A, const void* B, void* C,
                                    const void* bias, void* DER, int E, int M, int N,
                                    int K, int epi, hipStream_t stream) {
  switch (epi) {
    case EPI_NONE: launch_nt<EPI_NONE>(A, B, C, bias, DER, E, M, N, K, stream); break;
    case EPI_BIAS: launch_nt<EPI_BIAS>(A, B, C, bias, DER, E, M, N, K, stream); break;
    case EPI_BIAS_GELU:
      launch_nt<EPI_BIAS_GELU>(A, B, C, bias, DER, E, M, N, K, stream); break;
    case EPI_GELU_GRAD:
      launch_nt<EPI_GELU_GRAD>(A, B, C, bias, DER, E, M, N, K, stream); break;
    default: abort();
  }
}
