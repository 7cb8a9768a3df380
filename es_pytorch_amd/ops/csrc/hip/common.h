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

// fp8 (OCP e4m3fn on gfx950) pack/unpack via the native cvt instructions —
// encode and decode use the same hardware, so the round trip is
// self-consistent by construction.
typedef float es_f32x2 __attribute__((ext_vector_type(2)));

__device__ __forceinline__ void fp8x4_decode(uint32_t w, float* out4) {
  es_f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  es_f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
  out4[0] = lo.x;
  out4[1] = lo.y;
  out4[2] = hi.x;
  out4[3] = hi.y;
}

__device__ __forceinline__ uint32_t fp8x4_encode(float a, float b, float c, float d) {
  uint32_t w = 0;
  w = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, w, false);
  w = __builtin_amdgcn_cvt_pk_fp8_f32(c, d, w, true);
  return w;
}

__device__ __forceinline__ float fp8_byte(uint8_t b) {
  es_f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8((uint32_t)b, false);
  return lo.x;
}

// e4m3fn quantization round trip through the same hardware converters the
// pheno/rollout path uses — yields EXACTLY the value the rollout computed
__device__ __forceinline__ float e4m3_roundtrip(float v) {
  const uint32_t w = __builtin_amdgcn_cvt_pk_fp8_f32(v, v, 0u, false);
  es_f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  return lo.x;
}
