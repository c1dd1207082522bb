// Shared definitions for the _mzk CDNA4 kernel set (gfx950 / MI355X).
// Replaces the reference's znicz/ocl/defines.cl + per-kernel #define
// injection: everything is compile-time templated instead.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#include "api.h"

#define WAVE 64

// device side of api.h's DivM (see there for the exactness bound)
__device__ __forceinline__ unsigned divq(unsigned n, DivM m) {
  return (unsigned)(((unsigned long long)n * m.mul) >> m.shift);
}

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;

// ---------------------------------------------------------------------------
// bf16 <-> f32 (bit-level; RNE pack)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float bf2f(unsigned short h) {
  union { unsigned int u; float f; } v;
  v.u = ((unsigned int)h) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  union { unsigned int u; float f; } v;
  v.f = f;
  unsigned int lsb = (v.u >> 16) & 1u;
  unsigned int rounded = v.u + 0x7FFFu + lsb;   // RNE
  if ((v.u & 0x7F800000u) == 0x7F800000u) rounded = v.u;  // inf/nan keep
  return (unsigned short)(rounded >> 16);
}

// ---------------------------------------------------------------------------
// activations (codes must match veles/znicz_amd/activation_defs.py)
// ---------------------------------------------------------------------------
#define TANH_A 1.7159f
#define TANH_B 0.6666f
#define TANH_DA 1.143819f
#define TANH_DB 0.388484f

template <int A>
__device__ __forceinline__ float act_fwd(float x) {
  if (A == ACT_TANH) return TANH_A * tanhf(TANH_B * x);
  if (A == ACT_RELU) return x > 15.f ? x : log1pf(__expf(x));
  if (A == ACT_STRICT_RELU) return fmaxf(x, 0.f);
  if (A == ACT_SIGMOID) return 1.f / (1.f + __expf(-x));
  return x;
}

template <int A>
__device__ __forceinline__ float act_dy(float y) {
  if (A == ACT_TANH) return TANH_DA - TANH_DB * y * y;
  if (A == ACT_RELU) return 1.f - __expf(-y);
  if (A == ACT_STRICT_RELU) return y > 0.f ? 1.f : 0.f;
  if (A == ACT_SIGMOID) return y * (1.f - y);
  return 1.f;
}

// runtime-code dispatch helper
#define ACT_SWITCH(code, F)                                                  \
  switch (code) {                                                            \
    case ACT_TANH: F(ACT_TANH); break;                                       \
    case ACT_RELU: F(ACT_RELU); break;                                       \
    case ACT_STRICT_RELU: F(ACT_STRICT_RELU); break;                         \
    case ACT_SIGMOID: F(ACT_SIGMOID); break;                                 \
    default: F(ACT_LINEAR); break;                                           \
  }

// ---------------------------------------------------------------------------
// conv geometry (NHWC activations, [K][KH][KW][C] weights)
// ---------------------------------------------------------------------------
// ---------------------------------------------------------------------------
// stateless counter-based PRNG (splitmix64) for dropout / stochastic pool —
// the MI355X replacement for the reference's xorshift1024* state buffers
// (veles/ocl/random.cl): no state arrays, reproducible per (seed, step, i).
// ---------------------------------------------------------------------------
__device__ __forceinline__ float uniform01(unsigned long long seed,
                                           unsigned long long idx) {
  unsigned long long z = seed + idx * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);  // 24-bit mantissa
}

// fast x^(-beta) for LRN: the AlexNet beta = 3/4 hits the exact-
// instruction path (v_rsq + v_sqrt + v_mul) instead of exp2/log2
__device__ __forceinline__ float pow_nbeta(float d, float beta,
                                           bool b34) {
  if (b34) {
    float r = __builtin_amdgcn_rsqf(d);       // d^-1/2
    return r * __builtin_amdgcn_sqrtf(r);     // d^-3/4
  }
  return __powf(d, -beta);
}

// host-side: log2 of the pow2 lane->channel tile (≤256) for NHWC kernels:
// lanes cover channels with shifts/masks only — no per-element div/mod.
static inline int chan_lg(int C) {
  int lg = 0;
  while ((1 << lg) < C && lg < 8) ++lg;
  return lg;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e_ = hipGetLastError();                                       \
    if (e_ != hipSuccess)                                                    \
      throw std::runtime_error(hipGetErrorString(e_));                       \
  } while (0)
