// Causal flash attention (fwd + recompute bwd) for MI355X (gfx950).
//
// Round-2 perf design (replaces the round-1 LDS-staged correctness-first
// kernel; ladder in docs/KERNELS.md). Key decisions, each grounded in the
// CDNA4 guide's measurements at THIS shape class (S=1024, D=64, causal):
//
// 1. NO K/V LDS STAGING. At S=1024 one (batch, head)'s K/V is 256 KB and
//    L2/L3-resident; the guide measured LDS staging + transposed reads as
//    pure overhead here (cdna_hip_programming.md §5 common-mistake #7).
//    Every MFMA operand is loaded straight from global memory in exactly
//    the fragment shape the MFMA wants (bf16x8 = 16 B/lane, coalesced
//    across the 16-lane B-groups). Operands whose fragment would be a
//    strided column read (V in PV, K in dQ, Q/dO in dKV) are PRE-TRANSPOSED
//    once per call by the tiled transpose_bf16 kernel (optim.hip) so their
//    fragment reads are contiguous too.
// 2. BARRIER-FREE. One wave owns 32 query rows (fwd/dq) or 32 key rows
//    (dkv) end-to-end; the only LDS use is the wave-PRIVATE P/dS transpose
//    buffer (C-fragment layout -> A-fragment layout between the two MFMAs),
//    so there is no __syncthreads anywhere and each wave stops at its OWN
//    causal bound (wave 0 of a block does ~25% less work than wave 3 and
//    retires early instead of idling at a barrier).
// 3. exp2 SOFTMAX. Scores are scaled into log2 domain once
//    (scale2 = scale * log2 e) so every exponential is a single v_exp_f32
//    (exp2f) instead of libm expf; the running max and LSE are tracked in
//    log2 domain and converted once at the end (LSE stays natural-log on
//    the wire for the backward contract).
//
// Shapes: q/k/v [BH, S, 64] bf16, S % 128 == 0, causal, head_dim 64.
#include "common.h"
#include <cmath>

#define AT_LD 72        // P/dS LDS row stride (64 + 8: b128-read bank spread)
#define LOG2E 1.4426950408889634f
#define LN2 0.6931471805599453f
#define NEG_BIG (-3.0e38f)

// raw v_exp_f32 / v_log_f32 / v_rcp_f32: softmax arguments are <= 0 and
// l >= 1, so libm's denorm/range fixup code (~5 VALU per call) buys nothing
LAH_DEVINL float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }
LAH_DEVINL float fast_log2(float x) { return __builtin_amdgcn_logf(x); }
LAH_DEVINL float fast_rcp(float x) { return __builtin_amdgcn_rcpf(x); }

// defer-max threshold (log2 domain): skip the cross-lane max reduction and
// the O/l rescale while no lane's tile max exceeds the running max by more
// than this — P is then bounded by 2^8 instead of 1, which f32 accumulation
// absorbs (guide T13; forced-rescale numerics test in tests/test_ops_gpu.py)
#define DEFER_THR 8.0f

// XCD-aware block remap (guide T1): the dispatcher places block b on XCD
// b % 8, so a PLAIN (qb-fastest) grid scatters one (b,h)'s query blocks
// over all 8 per-XCD L2s and its K/V tiles are re-fetched from HBM by every
// block. Remapping ids so that id ≡ bh (mod 8) pins all of one head's query
// blocks to ONE XCD: its 256 KB of K/V stays L2-resident across them.
// Bijective iff BH % 8 == 0; callers fall back to the identity otherwise.
LAH_DEVINL void attn_block_id(int nqb, int BH, int& qb, long& bh) {
  const int id = blockIdx.x;
  if ((BH & 7) == 0) {
    const int t = id >> 3;
    qb = t % nqb;
    bh = (id & 7) + 8 * (long)(t / nqb);
  } else {
    qb = id % nqb;
    bh = id / nqb;
  }
}

// ---------------------------------------------------------------------------
// forward: one wave = 32 q rows; block = 4 waves = 128 q rows of one (b,h)
// ---------------------------------------------------------------------------
template <int MINW, bool PREFETCH, int FKV>
__launch_bounds__(256, MINW) __global__
void flash_attn_fwd_kernel(const __bf16* __restrict__ Q,   // [BH, S, 64]
                           const __bf16* __restrict__ K,   // [BH, S, 64]
                           const __bf16* __restrict__ Vt,  // [BH, 64, S]
                           __bf16* __restrict__ O,
                           float* __restrict__ LSE,  // [BH, S] or null
                           int BH, int S, float scale2) {
  constexpr int KV = FKV * 16;   // key-tile width (64 or 128)
  constexpr int PS_LD = KV + 8;  // wave-private P row stride
  __shared__ __attribute__((aligned(16))) __bf16 smem[4 * 32 * (KV + 8)];

  int qb;   // 128-row query block
  long bh;
  attn_block_id(S / 128, BH, qb, bh);
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8, r0 = (lane >> 4) * 4;
  const long base = bh * (long)S * 64;
  const long baseT = bh * (long)64 * S;  // Vt
  const int qw = qb * 128 + wave * 32;   // this wave's first q row
  __bf16* Ps = smem + wave * 32 * PS_LD;

  // Q fragments, resident for the whole kernel: aq[m][kk]
  bf16x8 aq[2][2];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      aq[m][kk] = *(const bf16x8*)(Q + base + (long)(qw + m * 16 + l15) * 64 + kk * 32 + lk8);

  f32x4 of[2][4];
  f32x4 lf[2];  // softmax denominator, accumulated as the P @ [V | 1]
                // ones-column fragment — no cross-lane sum reduction needed
  float m_run[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int f = 0; f < 4; ++f) of[m][f] = {0.f, 0.f, 0.f, 0.f};
    lf[m] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < 4; ++r) m_run[m][r] = NEG_BIG;
  }
  // ones-column B fragment: column n=0 of a 16-wide block is 1, rest 0
  bf16x8 bones;
#pragma unroll
  for (int j = 0; j < 8; ++j) bones[j] = (l15 == 0) ? (__bf16)1.0f : (__bf16)0.0f;

  const int kb_last = (qw + 31) / KV;  // this wave's causal bound

  // K fragments double-buffered in registers: tile kb+1's loads issue
  // BEFORE tile kb's softmax/PV, so their HBM/L2 latency hides under
  // ~1.5k cycles of compute (guide T14 async-STAGE, register form)
  bf16x8 bkA[2][FKV], bkB[2][FKV];
  auto load_k = [&](bf16x8(&dst)[2][FKV], int kb_) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int f = 0; f < FKV; ++f)
        dst[kk][f] = *(const bf16x8*)(K + base + (long)(kb_ * KV + f * 16 + l15) * 64 + kk * 32 + lk8);
  };

  auto tile = [&](const bf16x8(&bk)[2][FKV], int kb) {
    // ---- S = Q @ K^T ----------------------------------------------------
    f32x4 sf[2][FKV];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < FKV; ++f) sf[m][f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int f = 0; f < FKV; ++f) {
        sf[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[0][kk], bk[kk][f], sf[0][f], 0, 0, 0);
        sf[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[1][kk], bk[kk][f], sf[1][f], 0, 0, 0);
      }
    // ---- log2-domain scale + (boundary-only) causal mask + lane max ------
    float rmax[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) rmax[m][r] = NEG_BIG;
    const bool boundary = (kb * KV + KV - 1) > qw;  // wave-uniform
    if (boundary) {
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int f = 0; f < FKV; ++f)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int qg = qw + m * 16 + r0 + r;
            int kg = kb * KV + f * 16 + l15;
            sf[m][f][r] = (kg <= qg) ? sf[m][f][r] * scale2 : NEG_BIG;
            rmax[m][r] = fmaxf(rmax[m][r], sf[m][f][r]);
          }
    } else {
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int f = 0; f < FKV; ++f)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            sf[m][f][r] *= scale2;
            rmax[m][r] = fmaxf(rmax[m][r], sf[m][f][r]);
          }
    }
    // ---- defer-max (T13): cross-lane reduce + rescale only when some
    // lane's tile max beats the running max by > DEFER_THR ------------------
    bool grew = false;
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) grew |= rmax[m][r] > m_run[m][r] + DEFER_THR;
    if (__any(grew)) {
#pragma unroll
      for (int x = 1; x < 16; x <<= 1)
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            rmax[m][r] = fmaxf(rmax[m][r], __shfl_xor(rmax[m][r], x, 64));
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float m_new = fmaxf(m_run[m][r], rmax[m][r]);
          float corr = fast_exp2(m_run[m][r] - m_new);
          m_run[m][r] = m_new;
          lf[m][r] *= corr;
#pragma unroll
          for (int f = 0; f < 4; ++f) of[m][f][r] *= corr;
        }
    }
    // ---- P -> wave-private LDS (bounded by 2^DEFER_THR, f32-absorbed) ----
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < FKV; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          Ps[(m * 16 + r0 + r) * PS_LD + f * 16 + l15] =
              f2bf(fast_exp2(sf[m][f][r] - m_run[m][r]));
    // ---- [O | l] += P @ [V | 1]: A = P rows (LDS), B = Vt rows (global) --
#pragma unroll
    for (int kk = 0; kk < KV / 32; ++kk) {
      bf16x8 ap0 = *(const bf16x8*)(Ps + (0 * 16 + l15) * PS_LD + kk * 32 + lk8);
      bf16x8 ap1 = *(const bf16x8*)(Ps + (1 * 16 + l15) * PS_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bv = *(const bf16x8*)(Vt + baseT + (long)(f * 16 + l15) * S + kb * KV + kk * 32 + lk8);
        of[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap0, bv, of[0][f], 0, 0, 0);
        of[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap1, bv, of[1][f], 0, 0, 0);
      }
      lf[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap0, bones, lf[0], 0, 0, 0);
      lf[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap1, bones, lf[1], 0, 0, 0);
    }
  };

  if constexpr (PREFETCH) {
    load_k(bkA, 0);
    int kb = 0;
    while (true) {  // 2-stage unroll: static buffer naming, no runtime index
      if (kb + 1 <= kb_last) load_k(bkB, kb + 1);
      tile(bkA, kb);
      if (++kb > kb_last) break;
      if (kb + 1 <= kb_last) load_k(bkA, kb + 1);
      tile(bkB, kb);
      if (++kb > kb_last) break;
    }
  } else {
    for (int kb = 0; kb <= kb_last; ++kb) {
      load_k(bkA, kb);
      tile(bkA, kb);
    }
  }

  // ---- epilogue: broadcast l from column 0, O * (1/l), LSE --------------
  float l_run[2][4], inv_l[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_run[m][r] = __shfl(lf[m][r], lane & 48, 64);  // col-0 lane of the group
      inv_l[m][r] = fast_rcp(l_run[m][r]);
    }
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = qw + m * 16 + r0 + r;
        O[base + (long)row * 64 + f * 16 + l15] = f2bf(of[m][f][r] * inv_l[m][r]);
      }
  if (LSE != nullptr && l15 == 0) {
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = qw + m * 16 + r0 + r;
        LSE[bh * S + row] = (m_run[m][r] + fast_log2(l_run[m][r])) * LN2;
      }
  }
}

// ---------------------------------------------------------------------------
// backward dQ: one wave = 32 q rows; needs Kt [BH, 64, S]
//   dP = dO @ V^T (B = V rows), P = exp2(scale2*S - lse2),
//   dS = P * (dP - D) * scale, dQ = dS @ K (B = Kt rows)
// ---------------------------------------------------------------------------
__launch_bounds__(256, 2) __global__
void flash_attn_bwd_dq_kernel(const __bf16* __restrict__ Q, const __bf16* __restrict__ K,
                              const __bf16* __restrict__ V, const __bf16* __restrict__ Kt,
                              const __bf16* __restrict__ dO,
                              const float* __restrict__ LSE, const float* __restrict__ Dv,
                              __bf16* __restrict__ dQ, int BH, int S, float scale) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[4 * 32 * AT_LD];
  int qb;
  long bh;
  attn_block_id(S / 128, BH, qb, bh);
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8, r0 = (lane >> 4) * 4;
  const long base = bh * (long)S * 64;
  const long baseT = bh * (long)64 * S;
  const int qw = qb * 128 + wave * 32;
  const float scale2 = scale * LOG2E;
  __bf16* dSs = smem + wave * 32 * AT_LD;

  bf16x8 aq[2][2], ad[2][2];
  float lse2[2][4], d_r[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      aq[m][kk] = *(const bf16x8*)(Q + base + (long)(qw + m * 16 + l15) * 64 + kk * 32 + lk8);
      ad[m][kk] = *(const bf16x8*)(dO + base + (long)(qw + m * 16 + l15) * 64 + kk * 32 + lk8);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = qw + m * 16 + r0 + r;
      lse2[m][r] = LSE[bh * S + row] * LOG2E;
      d_r[m][r] = Dv[bh * S + row];
    }
  }
  f32x4 dq[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int f = 0; f < 4; ++f) dq[m][f] = {0.f, 0.f, 0.f, 0.f};

  const int kb_last = (qw + 31) >> 6;
  for (int kb = 0; kb <= kb_last; ++kb) {
    f32x4 sf[2][4], df[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < 4; ++f) { sf[m][f] = {0.f, 0.f, 0.f, 0.f}; df[m][f] = {0.f, 0.f, 0.f, 0.f}; }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bk = *(const bf16x8*)(K + base + (long)(kb * 64 + f * 16 + l15) * 64 + kk * 32 + lk8);
        bf16x8 bv = *(const bf16x8*)(V + base + (long)(kb * 64 + f * 16 + l15) * 64 + kk * 32 + lk8);
        sf[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[0][kk], bk, sf[0][f], 0, 0, 0);
        sf[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[1][kk], bk, sf[1][f], 0, 0, 0);
        df[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ad[0][kk], bv, df[0][f], 0, 0, 0);
        df[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ad[1][kk], bv, df[1][f], 0, 0, 0);
      }
    }
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < 4; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qg = qw + m * 16 + r0 + r;
          int kg = kb * 64 + f * 16 + l15;
          float p = (kg <= qg) ? fast_exp2(sf[m][f][r] * scale2 - lse2[m][r]) : 0.f;
          float ds = p * (df[m][f][r] - d_r[m][r]) * scale;
          dSs[(m * 16 + r0 + r) * AT_LD + f * 16 + l15] = f2bf(ds);
        }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 as0 = *(const bf16x8*)(dSs + (0 * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 as1 = *(const bf16x8*)(dSs + (1 * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bkt = *(const bf16x8*)(Kt + baseT + (long)(f * 16 + l15) * S + kb * 64 + kk * 32 + lk8);
        dq[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as0, bkt, dq[0][f], 0, 0, 0);
        dq[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as1, bkt, dq[1][f], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = qw + m * 16 + r0 + r;
        dQ[base + (long)row * 64 + f * 16 + l15] = f2bf(dq[m][f][r]);
      }
}

// ---------------------------------------------------------------------------
// backward dK/dV: one wave = 32 KEY rows; needs Qt, dOt [BH, 64, S]
//   S^T = K @ Q^T, dP^T = V @ dO^T  (B-frags = Q/dO rows),
//   dV = P^T @ dO (B = dOt rows), dK = dS^T @ Q (B = Qt rows)
// ---------------------------------------------------------------------------
__launch_bounds__(256, 2) __global__
void flash_attn_bwd_dkv_kernel(const __bf16* __restrict__ Q, const __bf16* __restrict__ K,
                               const __bf16* __restrict__ V, const __bf16* __restrict__ Qt,
                               const __bf16* __restrict__ dO, const __bf16* __restrict__ dOt,
                               const float* __restrict__ LSE, const float* __restrict__ Dv,
                               __bf16* __restrict__ dK, __bf16* __restrict__ dV,
                               int BH, int S, float scale) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[4 * 2 * 32 * AT_LD];
  int kbk;  // 128-row key block
  long bh;
  attn_block_id(S / 128, BH, kbk, bh);
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8, r0 = (lane >> 4) * 4;
  const long base = bh * (long)S * 64;
  const long baseT = bh * (long)64 * S;
  const int kw = kbk * 128 + wave * 32;  // this wave's first key row
  const float scale2 = scale * LOG2E;
  __bf16* Ps = smem + wave * 2 * 32 * AT_LD;
  __bf16* dSs = Ps + 32 * AT_LD;

  bf16x8 ak[2][2], av[2][2];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      ak[m][kk] = *(const bf16x8*)(K + base + (long)(kw + m * 16 + l15) * 64 + kk * 32 + lk8);
      av[m][kk] = *(const bf16x8*)(V + base + (long)(kw + m * 16 + l15) * 64 + kk * 32 + lk8);
    }
  f32x4 dk[2][4], dv[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int f = 0; f < 4; ++f) { dk[m][f] = {0.f, 0.f, 0.f, 0.f}; dv[m][f] = {0.f, 0.f, 0.f, 0.f}; }

  const int nqb = S / 64;
  for (int qt = kw / 64; qt < nqb; ++qt) {
    const int q0 = qt * 64;
    // per-column q stats (columns n = f*16 + l15 of this q tile)
    float lse2_c[4], d_c[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      lse2_c[f] = LSE[bh * S + q0 + f * 16 + l15] * LOG2E;
      d_c[f] = Dv[bh * S + q0 + f * 16 + l15];
    }
    f32x4 stf[2][4], dpf[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < 4; ++f) { stf[m][f] = {0.f, 0.f, 0.f, 0.f}; dpf[m][f] = {0.f, 0.f, 0.f, 0.f}; }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bq = *(const bf16x8*)(Q + base + (long)(q0 + f * 16 + l15) * 64 + kk * 32 + lk8);
        bf16x8 bd = *(const bf16x8*)(dO + base + (long)(q0 + f * 16 + l15) * 64 + kk * 32 + lk8);
        stf[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[0][kk], bq, stf[0][f], 0, 0, 0);
        stf[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[1][kk], bq, stf[1][f], 0, 0, 0);
        dpf[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[0][kk], bd, dpf[0][f], 0, 0, 0);
        dpf[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av[1][kk], bd, dpf[1][f], 0, 0, 0);
      }
    }
    const bool boundary = q0 < kw + 31;  // some (kg > qg) pairs in this tile
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int f = 0; f < 4; ++f)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int kg = kw + m * 16 + r0 + r;
          int qg = q0 + f * 16 + l15;
          float p = (!boundary || kg <= qg)
                        ? fast_exp2(stf[m][f][r] * scale2 - lse2_c[f]) : 0.f;
          float ds = p * (dpf[m][f][r] - d_c[f]) * scale;
          int lrow = (m * 16 + r0 + r) * AT_LD + f * 16 + l15;
          Ps[lrow] = f2bf(p);
          dSs[lrow] = f2bf(ds);
        }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 ap0 = *(const bf16x8*)(Ps + (0 * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 ap1 = *(const bf16x8*)(Ps + (1 * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 as0 = *(const bf16x8*)(dSs + (0 * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 as1 = *(const bf16x8*)(dSs + (1 * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bdt = *(const bf16x8*)(dOt + baseT + (long)(f * 16 + l15) * S + q0 + kk * 32 + lk8);
        bf16x8 bqt = *(const bf16x8*)(Qt + baseT + (long)(f * 16 + l15) * S + q0 + kk * 32 + lk8);
        dv[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap0, bdt, dv[0][f], 0, 0, 0);
        dv[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap1, bdt, dv[1][f], 0, 0, 0);
        dk[0][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as0, bqt, dk[0][f], 0, 0, 0);
        dk[1][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as1, bqt, dk[1][f], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = kw + m * 16 + r0 + r;
        dK[base + (long)row * 64 + f * 16 + l15] = f2bf(dk[m][f][r]);
        dV[base + (long)row * 64 + f * 16 + l15] = f2bf(dv[m][f][r]);
      }
}

extern "C" void lah_transpose_bf16(const void* in, void* out, int E, int R, int C,
                                   hipStream_t stream);

// Vt scratch is allocated by the caller (bindings.cpp) so allocation goes
// through torch's caching allocator, not raw hipMalloc.
extern "C" void lah_flash_attn_fwd(const void* Q, const void* K, const void* V,
                                   void* Vt_scratch, void* O, void* LSE,
                                   int BH, int S, hipStream_t stream) {
  lah_transpose_bf16(V, Vt_scratch, BH, S, 64, stream);
  dim3 grid((S / 128) * BH), block(256);
  const float scale2 = (1.0f / sqrtf(64.0f)) * LOG2E;
  // occupancy/prefetch variant A/B knob (perf work; default = measured best)
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("LAH_ATTN_V");
    v = e ? atoi(e) : 0;
  }
  auto kern = flash_attn_fwd_kernel<2, true, 4>;       // v0: 2 waves + K prefetch
  if (v == 1) kern = flash_attn_fwd_kernel<3, false, 4>;  // v1: 3 waves, no prefetch
  if (v == 2) kern = flash_attn_fwd_kernel<3, true, 4>;   // v2: 3 waves + prefetch (spills)
  if (v == 3) kern = flash_attn_fwd_kernel<2, false, 8>;  // v3: 128-wide KV tiles
  hipLaunchKernelGGL(kern, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)Vt_scratch,
                     (__bf16*)O, (float*)LSE, BH, S, scale2);
}

extern "C" void lah_flash_attn_bwd(const void* Q, const void* K, const void* V,
                                   const void* dO, const void* LSE, const void* Dv,
                                   void* Qt_s, void* Kt_s, void* dOt_s,
                                   void* dQ, void* dK, void* dV, int BH, int S,
                                   hipStream_t stream) {
  lah_transpose_bf16(Q, Qt_s, BH, S, 64, stream);
  lah_transpose_bf16(K, Kt_s, BH, S, 64, stream);
  lah_transpose_bf16(dO, dOt_s, BH, S, 64, stream);
  dim3 grid((S / 128) * BH), block(256);
  float scale = 1.0f / sqrtf(64.0f);
  hipLaunchKernelGGL(flash_attn_bwd_dq_kernel, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (const __bf16*)Kt_s, (const __bf16*)dO,
                     (const float*)LSE, (const float*)Dv, (__bf16*)dQ, BH, S, scale);
  hipLaunchKernelGGL(flash_attn_bwd_dkv_kernel, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (const __bf16*)Qt_s, (const __bf16*)dO, (const __bf16*)dOt_s,
                     (const float*)LSE, (const float*)Dv,
                     (__bf16*)dK, (__bf16*)dV, BH, S, scale);
}
