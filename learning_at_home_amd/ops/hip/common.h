// Common device helpers for learning_at_home_amd CDNA4 (gfx950) kernels.
// Written for MI355X only: wave64, MFMA bf16, LDS 160 KiB/CU, 8 XCDs.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define LAH_WAVE 64
#define LAH_DEVINL __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(2))) float f32x2;

LAH_DEVINL float bf2f(__bf16 x) { return (float)x; }
LAH_DEVINL __bf16 f2bf(float x) { return (__bf16)x; }

// tanh-approx GELU, matching torch.nn.functional.gelu(approximate="tanh")
#define GELU_C0 0.7978845608028654f  // sqrt(2/pi)
#define GELU_C1 0.044715f

LAH_DEVINL float gelu_fwd(float x) {
  float u = GELU_C0 * (x + GELU_C1 * x * x * x);
  return 0.5f * x * (1.0f + tanhf(u));
}

LAH_DEVINL float gelu_bwd(float x) {  // d gelu(x) / dx
  float x2 = x * x;
  float u = GELU_C0 * (x + GELU_C1 * x * x2);
  float t = tanhf(u);
  float du = GELU_C0 * (1.0f + 3.0f * GELU_C1 * x2);
  return 0.5f * (1.0f + t) + 0.5f * x * (1.0f - t * t) * du;
}

__host__ __device__ __forceinline__ int cdiv(int a, int b) { return (a + b - 1) / b; }

// async global->LDS copy, 16 bytes per lane (see cdna_hip_programming.md §5:
// the compiler never auto-emits global_load_lds; width 16 is the fast one)
LAH_DEVINL void glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}


// --- stochastic-rounding helpers (optim.hip SR-SGD + the fused TN+SGD
// weight-gradient kernel in ffn_gemm_tn.hip) ---------------------------------
LAH_DEVINL unsigned hash_u32(unsigned x) {
  x ^= x >> 16; x *= 0x7feb352du; x ^= x >> 15; x *= 0x846ca68bu; x ^= x >> 16;
  return x;
}

LAH_DEVINL __bf16 round_sr(float v, unsigned rnd16) {
  unsigned bits = __float_as_uint(v);
  unsigned frac = bits & 0xFFFFu;
  unsigned trunc = bits & 0xFFFF0000u;
  if (frac > (rnd16 & 0xFFFFu)) trunc += 0x10000u;
  unsigned short hi = (unsigned short)(trunc >> 16);
  return *reinterpret_cast<__bf16*>(&hi);
}

#define HIP_CHECK(expr)                                                              \
  do {                                                                               \
    hipError_t _e = (expr);                                                          \
    if (_e != hipSuccess) {                                                          \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
              __LINE__);                                                             \
      abort();                                                                       \
    }                                                                                \
  } while (0)
