// Causal flash-attention FORWARD for MI355X (gfx950) — experimental op
// (round-1 groundwork for replacing torch sdpa in TransformerExpert /
// TransformerDMoE inference; training still uses sdpa — no backward here).
//
// Correctness-first design, one workgroup = 4 waves = 64 query rows of one
// (batch, head); each wave owns 16 rows end-to-end (no cross-wave reduction):
//   for each 64-wide key block (causal upper bound):
//     S = Q @ K^T via mfma_f32_16x16x32_bf16        (frag maps as ffn_gemm.hip)
//     online softmax per row (shfl_xor row reduction across the 16 col lanes)
//     P -> LDS (bf16), O = O*corr + P @ V^T_t via MFMA (V staged transposed)
// Requires head_dim == 64, seq_len % 64 == 0, bf16, causal.
#include "common.h"
#include <cmath>

#define AT_LD 72  // LDS row stride in elements (64 + 8: 16B-aligned, depadded banks)

// Plain (unswizzled) tile stage: rows x 64 bf16, row-major with AT_LD stride.
template <int ROWS, int THREADS>
LAH_DEVINL void at_stage(const __bf16* __restrict__ src, __bf16* lds, long row0,
                         long ld, int tid) {
#pragma unroll
  for (int i = 0; i < (ROWS * 64) / (THREADS * 8); ++i) {
    int idx = i * THREADS + tid;  // bf16x8 chunk index, 8 chunks per row
    int row = idx >> 3, c8 = idx & 7;
    *(bf16x8*)(lds + row * AT_LD + c8 * 8) =
        *(const bf16x8*)(src + (row0 + row) * ld + c8 * 8);
  }
}

// Stage V TRANSPOSED: Vt[d][sk] = V[sk][d] (scalar LDS stores; correctness
// first — the P@V MFMA needs V^T as its [N, K] "B" operand).
template <int THREADS>
LAH_DEVINL void at_stage_vt(const __bf16* __restrict__ v, __bf16* lds, long row0,
                            long ld, int tid) {
#pragma unroll
  for (int i = 0; i < (64 * 64) / (THREADS * 8); ++i) {
    int idx = i * THREADS + tid;
    int sk = idx >> 3, d8 = (idx & 7) * 8;
    bf16x8 val = *(const bf16x8*)(v + (row0 + sk) * ld + d8);
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[(d8 + j) * AT_LD + sk] = val[j];
  }
}

__launch_bounds__(256, 2) __global__
void flash_attn_fwd_kernel(const __bf16* __restrict__ Q,  // [BH, S, 64]
                           const __bf16* __restrict__ K,
                           const __bf16* __restrict__ V,
                           __bf16* __restrict__ O,
                           float* __restrict__ LSE,  // [BH, S] or null (for bwd)
                           int S, float scale) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[4 * 64 * AT_LD];
  __bf16* Qs = smem;                // [64][AT_LD]
  __bf16* Ks = smem + 64 * AT_LD;
  __bf16* Vt = smem + 2 * 64 * AT_LD;  // [d][sk]
  __bf16* Ps = smem + 3 * 64 * AT_LD;  // wave w owns rows 16w..16w+15

  const int qb = blockIdx.x;            // query block (64 rows)
  const long bh = blockIdx.y;
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8;
  const long base = bh * (long)S * 64;
  const int q0 = qb * 64;

  at_stage<64, 256>(Q + base, Qs, q0, 64, tid);

  f32x4 Oacc[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) Oacc[f] = {0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }

  for (int kb = 0; kb <= qb; ++kb) {
    __syncthreads();  // previous iteration's MFMA reads of Ks/Vt are done
    at_stage<64, 256>(K + base, Ks, kb * 64, 64, tid);
    at_stage_vt<256>(V + base, Vt, kb * 64, 64, tid);
    __syncthreads();

    // ---- S = Q @ K^T (this wave's 16 rows x 64 cols) ------------------
    f32x4 sf[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) sf[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // Dh=64 -> two 32-wide k steps
      bf16x8 aq = *(const bf16x8*)(Qs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bk = *(const bf16x8*)(Ks + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        sf[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq, bk, sf[f], 0, 0, 0);
      }
    }
    // ---- scale + causal mask + online softmax -------------------------
    // C frag: col = l15 (+16f), row = (lane>>4)*4 + r (+16*wave), global +q0/kb*64
    float rmax[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qg = q0 + wave * 16 + (lane >> 4) * 4 + r;
        int kg = kb * 64 + f * 16 + l15;
        sf[f][r] = (kg <= qg) ? sf[f][r] * scale : -INFINITY;
        rmax[r] = fmaxf(rmax[r], sf[f][r]);
      }
#pragma unroll
    for (int x = 1; x < 16; x <<= 1)  // reduce across the 16 col lanes
#pragma unroll
      for (int r = 0; r < 4; ++r)
        rmax[r] = fmaxf(rmax[r], __shfl_xor(rmax[r], x, 64));
    float corr[4], rsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], rmax[r]);
      corr[r] = expf(m_run[r] - m_new);  // exp(-inf - finite) = 0 on round 0
      m_run[r] = m_new;
    }
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = expf(sf[f][r] - m_run[r]);  // masked: exp(-inf)=0
        sf[f][r] = p;
        rsum[r] += p;
      }
#pragma unroll
    for (int x = 1; x < 16; x <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r) rsum[r] += __shfl_xor(rsum[r], x, 64);
#pragma unroll
    for (int r = 0; r < 4; ++r) l_run[r] = l_run[r] * corr[r] + rsum[r];

    // P (bf16) to this wave's private LDS rows; rescale O accumulators
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        Ps[(wave * 16 + (lane >> 4) * 4 + r) * AT_LD + f * 16 + l15] = f2bf(sf[f][r]);
        Oacc[f][r] *= corr[r];
      }
    // ---- O += P @ V  (B operand = Vt[d][sk], NT semantics) ------------
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // Sk=64 -> two 32-wide k steps
      bf16x8 ap = *(const bf16x8*)(Ps + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bv = *(const bf16x8*)(Vt + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        Oacc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, Oacc[f], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l, scalar bf16 stores (correctness-first) --------
#pragma unroll
  for (int f = 0; f < 4; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
      O[base + (long)row * 64 + f * 16 + l15] = f2bf(Oacc[f][r] / l_run[r]);
    }
  if (LSE != nullptr && l15 == 0) {  // row stats are uniform across col lanes
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
      LSE[bh * S + row] = m_run[r] + logf(l_run[r]);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward, recompute-based (two kernels, no atomics):
//   dQ kernel  — grid over QUERY blocks; dQ[q] = scale * dS[q,:] @ K
//   dKV kernel — grid over KEY blocks;   dV[k] = P^T[k,:] @ dO,
//                                        dK[k] = scale * dS^T[k,:] @ Q
// with P = exp(scale*QK^T - LSE[q]) (recomputed from the saved LSE) and
// dS = P * (dO@V^T - D[q]), D[q] = rowsum(dO*O) precomputed by the wrapper.
// All products use the NT fragment maps proven in ffn_gemm.hip; operands
// that must enter MFMA as [N, K-dim] are staged transposed in LDS.
// ---------------------------------------------------------------------------

__launch_bounds__(256, 1) __global__
void flash_attn_bwd_dq_kernel(const __bf16* __restrict__ Q, const __bf16* __restrict__ K,
                              const __bf16* __restrict__ V, const __bf16* __restrict__ dO,
                              const float* __restrict__ LSE, const float* __restrict__ Dv,
                              __bf16* __restrict__ dQ, int S, float scale) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[6 * 64 * AT_LD];
  __bf16* Qs = smem;
  __bf16* dOs = smem + 64 * AT_LD;
  __bf16* Ks = smem + 2 * 64 * AT_LD;
  __bf16* Vs = smem + 3 * 64 * AT_LD;
  __bf16* Kt = smem + 4 * 64 * AT_LD;   // K transposed: Kt[d][sk]
  __bf16* dSs = smem + 5 * 64 * AT_LD;  // wave-private rows

  const int qb = blockIdx.x;
  const long bh = blockIdx.y;
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8;
  const long base = bh * (long)S * 64;
  const int q0 = qb * 64;

  at_stage<64, 256>(Q + base, Qs, q0, 64, tid);
  at_stage<64, 256>(dO + base, dOs, q0, 64, tid);
  float lse_r[4], d_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
    lse_r[r] = LSE[bh * S + row];
    d_r[r] = Dv[bh * S + row];
  }
  f32x4 dq[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) dq[f] = {0.f, 0.f, 0.f, 0.f};

  for (int kb = 0; kb <= qb; ++kb) {
    __syncthreads();
    at_stage<64, 256>(K + base, Ks, kb * 64, 64, tid);
    at_stage<64, 256>(V + base, Vs, kb * 64, 64, tid);
    at_stage_vt<256>(K + base, Kt, kb * 64, 64, tid);
    __syncthreads();

    f32x4 sf[4], df[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) { sf[f] = {0.f, 0.f, 0.f, 0.f}; df[f] = {0.f, 0.f, 0.f, 0.f}; }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 aq = *(const bf16x8*)(Qs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 ad = *(const bf16x8*)(dOs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bk = *(const bf16x8*)(Ks + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        bf16x8 bv = *(const bf16x8*)(Vs + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        sf[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq, bk, sf[f], 0, 0, 0);
        df[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ad, bv, df[f], 0, 0, 0);
      }
    }
    // dS = P * (dP - D) * scale, P = exp(scale*S - lse); causal mask -> 0
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qg = q0 + wave * 16 + (lane >> 4) * 4 + r;
        int kg = kb * 64 + f * 16 + l15;
        float p = (kg <= qg) ? expf(sf[f][r] * scale - lse_r[r]) : 0.f;
        float ds = p * (df[f][r] - d_r[r]) * scale;
        dSs[(wave * 16 + (lane >> 4) * 4 + r) * AT_LD + f * 16 + l15] = f2bf(ds);
      }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 as = *(const bf16x8*)(dSs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bkt = *(const bf16x8*)(Kt + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        dq[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as, bkt, dq[f], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int f = 0; f < 4; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
      dQ[base + (long)row * 64 + f * 16 + l15] = f2bf(dq[f][r]);
    }
}

__launch_bounds__(256, 1) __global__
void flash_attn_bwd_dkv_kernel(const __bf16* __restrict__ Q, const __bf16* __restrict__ K,
                               const __bf16* __restrict__ V, const __bf16* __restrict__ dO,
                               const float* __restrict__ LSE, const float* __restrict__ Dv,
                               __bf16* __restrict__ dK, __bf16* __restrict__ dV,
                               int S, float scale) {
  __shared__ __attribute__((aligned(16))) __bf16 smem[8 * 64 * AT_LD];
  __shared__ float lse_s[64], d_s[64];
  __bf16* Ks = smem;                    // this block's K rows (A operand)
  __bf16* Vs = smem + 64 * AT_LD;
  __bf16* Qs = smem + 2 * 64 * AT_LD;   // per-q-block tiles
  __bf16* dOs = smem + 3 * 64 * AT_LD;
  __bf16* Qt = smem + 4 * 64 * AT_LD;   // transposed: Qt[d][q]
  __bf16* dOt = smem + 5 * 64 * AT_LD;
  __bf16* Ps = smem + 6 * 64 * AT_LD;   // P^T rows (wave-private)
  __bf16* dSs = smem + 7 * 64 * AT_LD;  // dS^T rows (wave-private)

  const int kb = blockIdx.x;
  const long bh = blockIdx.y;
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int l15 = lane & 15, lk8 = (lane >> 4) * 8;
  const long base = bh * (long)S * 64;
  const int k0 = kb * 64;
  const int nqb = S / 64;

  at_stage<64, 256>(K + base, Ks, k0, 64, tid);
  at_stage<64, 256>(V + base, Vs, k0, 64, tid);
  f32x4 dk[4], dv[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) { dk[f] = {0.f, 0.f, 0.f, 0.f}; dv[f] = {0.f, 0.f, 0.f, 0.f}; }

  for (int qb = kb; qb < nqb; ++qb) {
    const int q0 = qb * 64;
    __syncthreads();
    at_stage<64, 256>(Q + base, Qs, q0, 64, tid);
    at_stage<64, 256>(dO + base, dOs, q0, 64, tid);
    at_stage_vt<256>(Q + base, Qt, q0, 64, tid);
    at_stage_vt<256>(dO + base, dOt, q0, 64, tid);
    if (tid < 64) {
      lse_s[tid] = LSE[bh * S + q0 + tid];
      d_s[tid] = Dv[bh * S + q0 + tid];
    }
    __syncthreads();

    // S^T = K @ Q^T and dP^T = V @ dO^T (C rows = k, cols = q)
    f32x4 stf[4], dpf[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) { stf[f] = {0.f, 0.f, 0.f, 0.f}; dpf[f] = {0.f, 0.f, 0.f, 0.f}; }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 ak = *(const bf16x8*)(Ks + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 av = *(const bf16x8*)(Vs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bq = *(const bf16x8*)(Qs + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        bf16x8 bd = *(const bf16x8*)(dOs + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        stf[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak, bq, stf[f], 0, 0, 0);
        dpf[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bd, dpf[f], 0, 0, 0);
      }
    }
    // P^T and dS^T into this wave's LDS rows
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kg = k0 + wave * 16 + (lane >> 4) * 4 + r;
        int qc = f * 16 + l15;  // column = query index within the block
        int qg = q0 + qc;
        float p = (kg <= qg) ? expf(stf[f][r] * scale - lse_s[qc]) : 0.f;
        float ds = p * (dpf[f][r] - d_s[qc]) * scale;
        int lrow = (wave * 16 + (lane >> 4) * 4 + r) * AT_LD + qc;
        Ps[lrow] = f2bf(p);
        dSs[lrow] = f2bf(ds);
      }
    // dV += P^T @ dO  (B = dOt[d][q]);  dK += dS^T @ Q  (B = Qt[d][q])
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 ap = *(const bf16x8*)(Ps + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
      bf16x8 as = *(const bf16x8*)(dSs + (wave * 16 + l15) * AT_LD + kk * 32 + lk8);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 bdt = *(const bf16x8*)(dOt + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        bf16x8 bqt = *(const bf16x8*)(Qt + (f * 16 + l15) * AT_LD + kk * 32 + lk8);
        dv[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bdt, dv[f], 0, 0, 0);
        dk[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as, bqt, dk[f], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int f = 0; f < 4; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = k0 + wave * 16 + (lane >> 4) * 4 + r;
      dK[base + (long)row * 64 + f * 16 + l15] = f2bf(dk[f][r]);
      dV[base + (long)row * 64 + f * 16 + l15] = f2bf(dv[f][r]);
    }
}

extern "C" void lah_flash_attn_fwd(const void* Q, const void* K, const void* V,
                                   void* O, void* LSE, int BH, int S,
                                   hipStream_t stream) {
  dim3 grid(S / 64, BH), block(256);
  hipLaunchKernelGGL(flash_attn_fwd_kernel, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (__bf16*)O, (float*)LSE, S, 1.0f / sqrtf(64.0f));
}

extern "C" void lah_flash_attn_bwd(const void* Q, const void* K, const void* V,
                                   const void* dO, const void* LSE, const void* Dv,
                                   void* dQ, void* dK, void* dV, int BH, int S,
                                   hipStream_t stream) {
  dim3 grid(S / 64, BH), block(256);
  float scale = 1.0f / sqrtf(64.0f);
  hipLaunchKernelGGL(flash_attn_bwd_dq_kernel, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (const __bf16*)dO, (const float*)LSE, (const float*)Dv,
                     (__bf16*)dQ, S, scale);
  hipLaunchKernelGGL(flash_attn_bwd_dkv_kernel, grid, block, 0, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (const __bf16*)dO, (const float*)LSE, (const float*)Dv,
                     (__bf16*)dK, (__bf16*)dV, S, scale);
}
