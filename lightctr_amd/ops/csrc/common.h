// Common device helpers for lightctr_amd CDNA4 (gfx950) kernels.
// Hand-written HIP for MI355X: 64-wide wavefronts, LDS, MFMA.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdio>

#define LCTR_WAVE 64

// Full-wave sum reduction (64 lanes).
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int s = 1; s < LCTR_WAVE; s <<= 1) v += __shfl_xor(v, s);
  return v;
}

// Reduce across stride groups: lanes {l, l+stride, l+2*stride, ...} end up
// holding the sum over their group (used to combine per-feature-group
// partials that live at lane = g*K + k for fixed k).
template <int START>
__device__ __forceinline__ float group_reduce_sum(float v) {
#pragma unroll
  for (int s = START; s < LCTR_WAVE; s <<= 1) v += __shfl_xor(v, s);
  return v;
}

__device__ __forceinline__ float sigmoidf_clamped(float z) {
  // Reference semantics: sigmoid clamped to +-16 pre-activation
  // (reference util/activations.h Sigmoid).
  z = fminf(16.f, fmaxf(-16.f, z));
  return 1.f / (1.f + __expf(-z));
}

// Runtime -> compile-time K dispatch (K must divide the 64-lane wave).
#define DISPATCH_K(KVAL, ...)                                        \
  switch (KVAL) {                                                    \
    case 4: { constexpr int KC = 4; __VA_ARGS__; break; }            \
    case 8: { constexpr int KC = 8; __VA_ARGS__; break; }            \
    case 16: { constexpr int KC = 16; __VA_ARGS__; break; }          \
    case 32: { constexpr int KC = 32; __VA_ARGS__; break; }          \
    case 64: { constexpr int KC = 64; __VA_ARGS__; break; }          \
    default:                                                         \
      fprintf(stderr, "lightctr_amd: unsupported K=%d\n", KVAL);     \
      abort();                                                       \
  }
#define DISPATCH_FFM_K DISPATCH_K

#define LCTR_CHECK_HIP(expr)                                        \
  do {                                                              \
    hipError_t _e = (expr);                                         \
    if (_e != hipSuccess) {                                         \
      fprintf(stderr, "HIP error %s at %s:%d\n",                    \
              hipGetErrorString(_e), __FILE__, __LINE__);           \
      abort();                                                      \
    }                                                               \
  } while (0)
