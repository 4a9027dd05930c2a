#include "hip/hip_runtime.h"
// Token dispatch / combine kernels for MI355X — the device side of the
// RCCL-all-to-all expert dispatch plane (SURVEY.md §2 "TaskPool" MI355X
// equivalent: capacity-bucketed expert batches; §2 hot-op #3: combine =
// softmax-over-survivors weighted sum, fwd + bwd).
//
// Layout: tokens [T, D] are routed to E experts with top-k slots; each
// expert has a CAPACITY-C block in the dispatch buffer [E, C, D]. Slots
// beyond capacity are DROPPED (pos = -1) — the reference's "drop
// non-responders, renormalize over survivors" semantics, applied at the
// capacity boundary; the combine renormalizes weights over surviving slots.
#include "common.h"

// ---- slot assignment -------------------------------------------------------
__global__ void assign_slots_kernel(const int* __restrict__ topk_idx,  // [T, k]
                                    int* __restrict__ pos,             // [T, k]
                                    int* __restrict__ counts,          // [E] zeroed
                                    int* __restrict__ inv,             // [E, C] (-1 fill)
                                    int total, int C, int E) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  int e = topk_idx[i];
  if (e < 0 || e >= E) {  // defensive: corrupt index must not become OOB
    pos[i] = -1;
    return;
  }
  int p = atomicAdd(&counts[e], 1);
  if (p < C) {
    pos[i] = p;
    inv[(long)e * C + p] = i;  // inverse map: (expert, slot) -> t*k + j
  } else {
    pos[i] = -1;  // capacity overflow: dropped
  }
}

// ---- dispatch gather: xbuf[e, c] = x[inv[e,c] / k], zeros for padding ------
__global__ void dispatch_gather_kernel(const __bf16* __restrict__ x,  // [T, D]
                                       const int* __restrict__ inv,   // [E, C]
                                       __bf16* __restrict__ xbuf,     // [E, C, D]
                                       int EC, int D, int k) {
  int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= EC) return;
  int lane = threadIdx.x & 63;
  int src = inv[row];
  __bf16* dst = xbuf + (long)row * D;
  if (src < 0) {
    for (int c = lane * 8; c < D; c += 64 * 8) *(bf16x8*)(dst + c) = bf16x8{};
  } else {
    const __bf16* s = x + (long)(src / k) * D;
    for (int c = lane * 8; c < D; c += 64 * 8) *(bf16x8*)(dst + c) = *(const bf16x8*)(s + c);
  }
}

// ---- dispatch backward: dx[t] = sum_j dxbuf[e_j, pos_j] --------------------
__global__ void dispatch_bwd_kernel(const __bf16* __restrict__ dxbuf,  // [E, C, D]
                                    const int* __restrict__ topk_idx,  // [T, k]
                                    const int* __restrict__ pos,       // [T, k]
                                    __bf16* __restrict__ dx,           // [T, D]
                                    int T, int D, int k, int C) {
  int t = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (t >= T) return;
  int lane = threadIdx.x & 63;
  for (int c = lane * 8; c < D; c += 64 * 8) {
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int j = 0; j < k; ++j) {
      int p = pos[(long)t * k + j];
      if (p < 0) continue;
      int e = topk_idx[(long)t * k + j];
      bf16x8 v = *(const bf16x8*)(dxbuf + ((long)e * C + p) * D + c);
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[q] += bf2f(v[q]);
    }
    bf16x8 out;
#pragma unroll
    for (int q = 0; q < 8; ++q) out[q] = f2bf(acc[q]);
    *(bf16x8*)(dx + (long)t * D + c) = out;
  }
}

// ---- combine forward: y[t] = sum_j (w_j / s) * ybuf[e_j, pos_j] ------------
__global__ void combine_fwd_kernel(const __bf16* __restrict__ ybuf,   // [E, C, D]
                                   const int* __restrict__ topk_idx,  // [T, k]
                                   const int* __restrict__ pos,       // [T, k]
                                   const float* __restrict__ w,       // [T, k]
                                   __bf16* __restrict__ y,            // [T, D]
                                   int T, int D, int k, int C) {
  int t = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (t >= T) return;
  int lane = threadIdx.x & 63;
  float s = 0.f;
  for (int j = 0; j < k; ++j)
    if (pos[(long)t * k + j] >= 0) s += w[(long)t * k + j];
  // clamp the renormalization mass: when capacity drops remove most of the
  // softmax mass, an unclamped 1/s amplifies combine gradients by up to
  // 1/w_min and detonates the router (observed at flagship scale)
  float inv_s = (s > 0.f) ? 1.0f / fmaxf(s, 0.2f) : 0.f;
  for (int c = lane * 8; c < D; c += 64 * 8) {
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int j = 0; j < k; ++j) {
      int p = pos[(long)t * k + j];
      if (p < 0) continue;
      int e = topk_idx[(long)t * k + j];
      float wj = w[(long)t * k + j] * inv_s;
      bf16x8 v = *(const bf16x8*)(ybuf + ((long)e * C + p) * D + c);
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[q] += wj * bf2f(v[q]);
    }
    bf16x8 out;
#pragma unroll
    for (int q = 0; q < 8; ++q) out[q] = f2bf(acc[q]);
    *(bf16x8*)(y + (long)t * D + c) = out;
  }
}

// ---- combine backward ------------------------------------------------------
// dybuf[e_j, pos_j] = (w_j/s) * dy[t]          (dybuf pre-zeroed for padding)
// d_w_j = (<dy, y_j> - <dy, y>) / s            (0 for dropped slots)
__global__ void combine_bwd_kernel(const __bf16* __restrict__ dy,     // [T, D]
                                   const __bf16* __restrict__ ybuf,   // [E, C, D]
                                   const __bf16* __restrict__ y,      // [T, D] (fwd out)
                                   const int* __restrict__ topk_idx,  // [T, k]
                                   const int* __restrict__ pos,       // [T, k]
                                   const float* __restrict__ w,       // [T, k]
                                   __bf16* __restrict__ dybuf,        // [E, C, D]
                                   float* __restrict__ dw,            // [T, k]
                                   int T, int D, int k, int C) {
  int t = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (t >= T) return;
  int lane = threadIdx.x & 63;
  float s = 0.f;
  for (int j = 0; j < k; ++j)
    if (pos[(long)t * k + j] >= 0) s += w[(long)t * k + j];
  float inv_s = (s > 0.f) ? 1.0f / fmaxf(s, 0.2f) : 0.f;  // see combine_fwd
  bool clamped = (s > 0.f) && (s < 0.2f);  // d(w/0.2)/dw has no quotient term

  // <dy, y> once per token
  float dot_dy_y = 0.f;
  for (int c = lane * 8; c < D; c += 64 * 8) {
    bf16x8 a = *(const bf16x8*)(dy + (long)t * D + c);
    bf16x8 b = *(const bf16x8*)(y + (long)t * D + c);
#pragma unroll
    for (int q = 0; q < 8; ++q) dot_dy_y += bf2f(a[q]) * bf2f(b[q]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) dot_dy_y += __shfl_xor(dot_dy_y, off);

  for (int j = 0; j < k; ++j) {
    int p = pos[(long)t * k + j];
    if (p < 0) {
      if (lane == 0) dw[(long)t * k + j] = 0.f;
      continue;
    }
    int e = topk_idx[(long)t * k + j];
    float wj = w[(long)t * k + j] * inv_s;
    long boff = ((long)e * C + p) * D;
    float dot_dy_yj = 0.f;
    for (int c = lane * 8; c < D; c += 64 * 8) {
      bf16x8 a = *(const bf16x8*)(dy + (long)t * D + c);
      bf16x8 b = *(const bf16x8*)(ybuf + boff + c);
      bf16x8 out;
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        dot_dy_yj += bf2f(a[q]) * bf2f(b[q]);
        out[q] = f2bf(wj * bf2f(a[q]));
      }
      *(bf16x8*)(dybuf + boff + c) = out;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) dot_dy_yj += __shfl_xor(dot_dy_yj, off);
    if (lane == 0)
      dw[(long)t * k + j] = (clamped ? dot_dy_yj : (dot_dy_yj - dot_dy_y)) * inv_s;
  }
}

// ---- zero-fill padding rows of a [E, C, D] buffer (rows with inv<0) --------
__global__ void zero_padding_kernel(const int* __restrict__ inv, __bf16* __restrict__ buf,
                                    int EC, int D) {
  int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= EC) return;
  if (inv[row] >= 0) return;
  int lane = threadIdx.x & 63;
  __bf16* dst = buf + (long)row * D;
  for (int c = lane * 8; c < D; c += 64 * 8) *(bf16x8*)(dst + c) = bf16x8{};
}

// ---- launchers -------------------------------------------------------------
extern "C" void lah_assign_slots(const void* topk_idx, void* pos, void* counts,
                                 void* inv, int T, int k, int C, int E,
                                 hipStream_t stream) {
  int total = T * k;
  hipLaunchKernelGGL(assign_slots_kernel, dim3(cdiv(total, 256)), dim3(256), 0, stream,
                     (const int*)topk_idx, (int*)pos, (int*)counts, (int*)inv, total, C, E);
}

extern "C" void lah_dispatch_gather(const void* x, const void* inv, void* xbuf, int E,
                                    int C, int D, int k, hipStream_t stream) {
  int rows = E * C;
  hipLaunchKernelGGL(dispatch_gather_kernel, dim3(cdiv(rows, 4)), dim3(256), 0, stream,
                     (const __bf16*)x, (const int*)inv, (__bf16*)xbuf, rows, D, k);
}

extern "C" void lah_dispatch_bwd(const void* dxbuf, const void* topk_idx, const void* pos,
                                 void* dx, int T, int D, int k, int C, hipStream_t stream) {
  hipLaunchKernelGGL(dispatch_bwd_kernel, dim3(cdiv(T, 4)), dim3(256), 0, stream,
                     (const __bf16*)dxbuf, (const int*)topk_idx, (const int*)pos,
                     (__bf16*)dx, T, D, k, C);
}

extern "C" void lah_combine_fwd(const void* ybuf, const void* topk_idx, const void* pos,
                                const void* w, void* y, int T, int D, int k, int C,
                                hipStream_t stream) {
  hipLaunchKernelGGL(combine_fwd_kernel, dim3(cdiv(T, 4)), dim3(256), 0, stream,
                     (const __bf16*)ybuf, (const int*)topk_idx, (const int*)pos,
                     (const float*)w, (__bf16*)y, T, D, k, C);
}

extern "C" void lah_combine_bwd(const void* dy, const void* ybuf, const void* y,
                                const void* topk_idx, const void* pos, const void* w,
                                void* dybuf, void* dw, int T, int D, int k, int C,
                                hipStream_t stream) {
  hipLaunchKernelGGL(combine_bwd_kernel, dim3(cdiv(T, 4)), dim3(256), 0, stream,
                     (const __bf16*)dy, (const __bf16*)ybuf, (const __bf16*)y,
                     (const int*)topk_idx, (const int*)pos, (const float*)w,
                     (__bf16*)dybuf, (float*)dw, T, D, k, C);
}

extern "C" void lah_zero_padding(const void* inv, void* buf, int E, int C, int D,
                                 hipStream_t stream) {
  int rows = E * C;
  hipLaunchKernelGGL(zero_padding_kernel, dim3(cdiv(rows, 4)), dim3(256), 0, stream,
                     (const int*)inv, (__bf16*)buf, rows, D);
}
