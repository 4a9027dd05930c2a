// Fused cross-entropy row kernels for the chunked lm_head CE
// (ops/fused_ce.py). The torch path per chunk materializes log-softmax /
// softmax over [T_chunk, 32000] fp32 several times (cunn_SoftMax* +
// elementwise ≈ 15 ms/step in the flagship profile); these two kernels make
// CE a single read of the logits per pass:
//   fwd:  lse[row] = log-sum-exp (online, exp2 domain), picked[row] =
//         logits[label] (= lse for ignored rows, so lse - picked == 0)
//   bwd:  dlogits = (softmax - onehot) * gscale, bf16, zero for ignored rows
// The lse vector ([T] fp32) is saved by the autograd ctx, so the backward
// needs no max/sum recompute — one GEMM + one elementwise pass per chunk.
//
// Templated on the logits dtype: the training path feeds the lm_head GEMM's
// bf16 output DIRECTLY (reading bf16 here is numerics-identical to casting
// to fp32 first — the values are already bf16-rounded — and skips both the
// 2×[T,V] fp32 materialization and its cast kernel, ~11 ms/step at the
// flagship shape); fp32 logits remain supported for fp32 callers.
#include "common.h"

#define CE_LOG2E 1.4426950408889634f
#define CE_LN2 0.6931471805599453f

template <typename TS, typename T4>
__launch_bounds__(256) __global__
void ce_rowstats_kernel(const TS* __restrict__ logits, const int* __restrict__ labels,
                        float* __restrict__ lse, float* __restrict__ picked,
                        int ignore_index, long V) {
  const int row = blockIdx.x;
  const TS* lr = logits + (long)row * V;
  const int tid = threadIdx.x;
  __shared__ float sm[8], ss[8];

  // per-thread online max/sum in log2 domain over a strided 4-wide sweep
  float m = -3.0e38f, s = 0.f;
  for (long c = (long)tid * 4; c + 3 < V; c += 256 * 4) {
    T4 v = *(const T4*)(lr + c);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float l2 = (float)v[j] * CE_LOG2E;
      if (l2 > m) {
        s = s * __builtin_amdgcn_exp2f(m - l2) + 1.0f;
        m = l2;
      } else {
        s += __builtin_amdgcn_exp2f(l2 - m);
      }
    }
  }
  // wave reduce (pairwise (m,s) combine), then cross-wave via LDS
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_xor(m, off, 64);
    float s2 = __shfl_xor(s, off, 64);
    float mn = fmaxf(m, m2);
    s = s * __builtin_amdgcn_exp2f(m - mn) + s2 * __builtin_amdgcn_exp2f(m2 - mn);
    m = mn;
  }
  const int wave = tid >> 6;
  if ((tid & 63) == 0) { sm[wave] = m; ss[wave] = s; }
  __syncthreads();
  if (tid == 0) {
    float mt = sm[0], st = ss[0];
#pragma unroll
    for (int w = 1; w < 4; ++w) {
      float mn = fmaxf(mt, sm[w]);
      st = st * __builtin_amdgcn_exp2f(mt - mn) + ss[w] * __builtin_amdgcn_exp2f(sm[w] - mn);
      mt = mn;
    }
    float l = (mt + __builtin_amdgcn_logf(st)) * CE_LN2;  // natural log
    lse[row] = l;
    int lab = labels[row];
    picked[row] = (lab == ignore_index || lab < 0 || lab >= V) ? l : (float)lr[lab];
  }
}

template <typename TS, typename T4>
__launch_bounds__(256) __global__
void ce_bwd_kernel(const TS* __restrict__ logits, const int* __restrict__ labels,
                   const float* __restrict__ lse, __bf16* __restrict__ dlogits,
                   float gscale, int ignore_index, long V) {
  const int row = blockIdx.y;
  const long c0 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (c0 >= V) return;
  const int lab = labels[row];
  const bool ignored = (lab == ignore_index || lab < 0);
  const float l2 = lse[row] * CE_LOG2E;
  const TS* lr = logits + (long)row * V;
  __bf16* dr = dlogits + (long)row * V;
  if (c0 + 3 < V) {
    T4 v = *(const T4*)(lr + c0);
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float p = ignored ? 0.f
                        : __builtin_amdgcn_exp2f((float)v[j] * CE_LOG2E - l2) * gscale
                              - ((c0 + j == lab) ? gscale : 0.f);
      o[j] = f2bf(p);
    }
    *(bf16x4*)(dr + c0) = o;
  } else {
    for (long c = c0; c < V; ++c) {
      float p = ignored ? 0.f
                        : __builtin_amdgcn_exp2f((float)lr[c] * CE_LOG2E - l2) * gscale
                              - ((c == lab) ? gscale : 0.f);
      dr[c] = f2bf(p);
    }
  }
}

extern "C" void lah_ce_rowstats(const void* logits, const void* labels, void* lse,
                                void* picked, int T, long V, int ignore_index,
                                int logits_bf16, hipStream_t stream) {
  if (logits_bf16)
    hipLaunchKernelGGL((ce_rowstats_kernel<__bf16, bf16x4>), dim3(T), dim3(256), 0,
                       stream, (const __bf16*)logits, (const int*)labels,
                       (float*)lse, (float*)picked, ignore_index, V);
  else
    hipLaunchKernelGGL((ce_rowstats_kernel<float, f32x4>), dim3(T), dim3(256), 0,
                       stream, (const float*)logits, (const int*)labels,
                       (float*)lse, (float*)picked, ignore_index, V);
}

extern "C" void lah_ce_bwd(const void* logits, const void* labels, const void* lse,
                           void* dlogits, float gscale, int T, long V,
                           int ignore_index, int logits_bf16, hipStream_t stream) {
  dim3 grid((unsigned)((V + 1023) / 1024), T), block(256);
  if (logits_bf16)
    hipLaunchKernelGGL((ce_bwd_kernel<__bf16, bf16x4>), grid, block, 0, stream,
                       (const __bf16*)logits, (const int*)labels, (const float*)lse,
                       (__bf16*)dlogits, gscale, ignore_index, V);
  else
    hipLaunchKernelGGL((ce_bwd_kernel<float, f32x4>), grid, block, 0, stream,
                       (const float*)logits, (const int*)labels, (const float*)lse,
                       (__bf16*)dlogits, gscale, ignore_index, V);
}
