// Shared helpers for the gfx950 ES kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#include <algorithm>

#include "../philox.h"

#define ES_CHECK_LAUNCH()                          \
  do {                                             \
    hipError_t e_ = hipGetLastError();             \
    if (e_ != hipSuccess) return (int)e_;          \
  } while (0)

// bf16 <-> f32 without header deps: round-to-nearest-even.
__device__ __forceinline__ float bf2f(uint16_t h) {
  union { uint32_t u; float f; } c;
  c.u = ((uint32_t)h) << 16;
  return c.f;
}

__device__ __forceinline__ uint16_t f2bf(float f) {
  union { uint32_t u; float f; } c;
  c.f = f;
  uint32_t u = c.u;
  uint32_t rounding = 0x7FFFu + ((u >> 16) & 1u);
  u += rounding;
  return (uint16_t)(u >> 16);
}

__device__ __forceinline__ float fclampf(float v, float lo, float hi) {
  return fminf(fmaxf(v, lo), hi);
}
