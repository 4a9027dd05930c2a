// Fused top-k gating kernels for MI355X (SURVEY.md §2 hot-op #2/#4:
// gating softmax + top-k + load-balancing loss as one fused pass).
//
// One 64-lane wave owns one token row (E <= 1024 experts, <=16 entries per
// lane, all in registers — no LDS round trip, cross-lane via __shfl_xor):
//   fwd: full softmax probs (bf16, for aux loss + backward), iterative
//        top-k argmax, softmax-over-selected weights.
//   bwd: d_logits = softmax-jacobian over the selected set (combine grads)
//        + load-balance auxiliary term p*(g_e - <g,p>) in the same pass.
#include "common.h"

#define GATE_WAVES 4  // waves (token rows) per block

LAH_DEVINL float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}
LAH_DEVINL float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

template <int EPL, int KT>  // entries per lane = ceil(E/64); KT = exact top-k
__global__ void gating_topk_fwd_kernel(const float* __restrict__ logits,  // [T, E]
                                       __bf16* __restrict__ probs,        // [T, E]
                                       int* __restrict__ topk_idx,        // [T, KT]
                                       float* __restrict__ topk_w,        // [T, KT]
                                       int T, int E) {
  constexpr int k = KT;
  int row = blockIdx.x * GATE_WAVES + (threadIdx.x >> 6);
  if (row >= T) return;
  int lane = threadIdx.x & 63;
  const float* lrow = logits + (long)row * E;

  float lv[EPL];
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int e = lane + i * 64;
    lv[i] = (e < E) ? lrow[e] : -INFINITY;
  }
  // full softmax
  float m = -INFINITY;
#pragma unroll
  for (int i = 0; i < EPL; ++i) m = fmaxf(m, lv[i]);
  m = wave_max(m);
  float s = 0.f;
  float ev[EPL];
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    ev[i] = (lv[i] == -INFINITY) ? 0.f : __expf(lv[i] - m);
    s += ev[i];
  }
  s = wave_sum(s);
  float inv_s = 1.0f / s;
  __bf16* prow = probs + (long)row * E;
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int e = lane + i * 64;
    if (e < E) prow[e] = f2bf(ev[i] * inv_s);
  }

  // iterative top-k (exact-k unrolled so sel_* stay in registers — scratch
  // otherwise, cdna_hip_programming.md §5.4 rule 20)
  unsigned sel_mask = 0;
  float sel_logit[KT];
  int sel_idx[KT];
  float msel = -INFINITY;
#pragma unroll
  for (int j = 0; j < k; ++j) {
    float best = -INFINITY;
    int bi = -1;
#pragma unroll
    for (int i = 0; i < EPL; ++i) {
      bool taken = (sel_mask >> i) & 1u;
      // NaN-safe: a NaN logit never wins, and bi falls back to the first
      // untaken entry so sel_idx is ALWAYS a valid expert (a negative index
      // here would turn into an out-of-bounds atomic in assign_slots)
      if (!taken && (bi < 0 || lv[i] > best)) { best = lv[i]; bi = i; }
    }
    if (bi < 0) bi = 0;
    // argmax across lanes: pack (value, lane) and reduce
    float bv = best;
    int bl = lane;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(bv, off);
      int ol = __shfl_xor(bl, off);
      if (ov > bv || (ov == bv && ol < bl)) { bv = ov; bl = ol; }
    }
    int bi_win = __shfl(bi, bl);  // winning lane's local entry index
    if (lane == bl) sel_mask |= (1u << bi);
    sel_logit[j] = bv;
    sel_idx[j] = bl + bi_win * 64;
    msel = fmaxf(msel, bv);
  }
  // softmax over the selected logits
  float wsum = 0.f;
#pragma unroll
  for (int j = 0; j < k; ++j) {
    sel_logit[j] = __expf(sel_logit[j] - msel);
    wsum += sel_logit[j];
  }
  if (lane == 0) {
#pragma unroll
    for (int j = 0; j < k; ++j) {
      topk_idx[(long)row * k + j] = sel_idx[j];
      topk_w[(long)row * k + j] = sel_logit[j] / wsum;
    }
  }
}

template <int EPL>
__global__ void gating_topk_bwd_kernel(const __bf16* __restrict__ probs,   // [T, E]
                                       const int* __restrict__ topk_idx,   // [T, k]
                                       const float* __restrict__ topk_w,   // [T, k]
                                       const float* __restrict__ d_topk_w, // [T, k]
                                       const float* __restrict__ aux_g,    // [E] or null
                                       float* __restrict__ d_logits,       // [T, E]
                                       int T, int E, int k) {
  int row = blockIdx.x * GATE_WAVES + (threadIdx.x >> 6);
  if (row >= T) return;
  int lane = threadIdx.x & 63;
  const __bf16* prow = probs + (long)row * E;
  float* drow = d_logits + (long)row * E;

  // aux-loss part: dl[e] = p_e * (g_e - <g, p>)  (zero if aux_g == null)
  float pv[EPL], gv[EPL];
  float dot = 0.f;
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int e = lane + i * 64;
    pv[i] = (e < E) ? bf2f(prow[e]) : 0.f;
    gv[i] = (aux_g != nullptr && e < E) ? aux_g[e] : 0.f;
    dot += pv[i] * gv[i];
  }
  dot = wave_sum(dot);
  float dl[EPL];
#pragma unroll
  for (int i = 0; i < EPL; ++i) dl[i] = pv[i] * (gv[i] - dot);

  // combine part: softmax-over-selected jacobian. Each lane loads slot `lane`
  // (if lane < k), the slot gradients are then broadcast to every lane via
  // shfl — no atomics, no barrier (safe with early-returned tail waves).
  float wj = 0.f, dj = 0.f;
  int ej = -1;
  if (lane < k) {
    wj = topk_w[(long)row * k + lane];
    dj = d_topk_w[(long)row * k + lane];
    ej = topk_idx[(long)row * k + lane];
  }
  float inner = wave_sum(dj * wj);  // sum_j d_j * w_j
  for (int j = 0; j < k; ++j) {
    int e_j = __shfl(ej, j);
    float g_j = __shfl(wj * (dj - inner), j);
#pragma unroll
    for (int i = 0; i < EPL; ++i) {
      if (lane + i * 64 == e_j) dl[i] += g_j;
    }
  }
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int e = lane + i * 64;
    if (e < E) drow[e] = dl[i];
  }
}

extern "C" void lah_gating_topk_fwd(const void* logits, void* probs, void* topk_idx,
                                    void* topk_w, int T, int E, int k,
                                    hipStream_t stream) {
  dim3 grid(cdiv(T, GATE_WAVES)), block(GATE_WAVES * 64);
  int epl = cdiv(E, 64);
#define CASE_K(N, K)                                                                   \
  if (epl <= N && k == K) {                                                            \
    hipLaunchKernelGGL((gating_topk_fwd_kernel<N, K>), grid, block, 0, stream,         \
                       (const float*)logits, (__bf16*)probs, (int*)topk_idx,           \
                       (float*)topk_w, T, E);                                          \
    return;                                                                            \
  }
#define CASE(N) CASE_K(N, 1) CASE_K(N, 2) CASE_K(N, 4) CASE_K(N, 8)
  // EPL 32/64 lift the wall to E <= 4096: 2*EPL VGPRs of
  // per-lane logit state, so occupancy drops for the huge-E tail — priced
  // only when such a config is actually run
  CASE(1) CASE(2) CASE(4) CASE(8) CASE(16) CASE(32) CASE(64)
#undef CASE
#undef CASE_K
  abort();  // E > 4096 or k not in {1,2,4,8}: unsupported (bindings TORCH_CHECK)
}

extern "C" void lah_gating_topk_bwd(const void* probs, const void* topk_idx,
                                    const void* topk_w, const void* d_topk_w,
                                    const void* aux_g, void* d_logits, int T, int E,
                                    int k, hipStream_t stream) {
  dim3 grid(cdiv(T, GATE_WAVES)), block(GATE_WAVES * 64);
  int epl = cdiv(E, 64);
#define CASE(N)                                                                        \
  if (epl <= N) {                                                                      \
    hipLaunchKernelGGL(gating_topk_bwd_kernel<N>, grid, block, 0, stream,              \
                       (const __bf16*)probs, (const int*)topk_idx,                     \
                       (const float*)topk_w, (const float*)d_topk_w,                   \
                       (const float*)aux_g, (float*)d_logits, T, E, k);                \
    return;                                                                            \
  }
  CASE(1) CASE(2) CASE(4) CASE(8) CASE(16) CASE(32) CASE(64)
#undef CASE
  abort();
}
