// Philox4x32-10 counter-based RNG + Box-Muller normal generation.
//
// Shared by the CPU extension (cpu_ops.cpp, g++) and the HIP kernels
// (hip/noise.hip, hipcc) so that the noise table generated on host and on
// device from the same seed is element-for-element identical up to libm/ocml
// ULP differences in (log, sqrt, sincos).
//
// This replaces the reference's node-shared MPI window filled by
// numpy RandomState.randn (reference src/core/noisetable.py:13-24,61-64,88):
// here every GPU fills its own HBM-resident replica deterministically from a
// broadcast 64-bit seed, so no host window and no
// rank-to-rank noise traffic exist at all.
#pragma once
#include <stdint.h>
#include <math.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define ES_HD __host__ __device__ __forceinline__
#else
#define ES_HD static inline
#endif

namespace esrng {

ES_HD uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hip) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hip = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

struct u32x4 { uint32_t x, y, z, w; };

// philox4x32-10: counter (c0..c3), key (k0,k1)
ES_HD u32x4 philox4x32(uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3,
                       uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(M0, c0, &hi0);
    uint32_t lo1 = mulhilo(M1, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  u32x4 out; out.x = c0; out.y = c1; out.z = c2; out.w = c3;
  return out;
}

// uniform in (0, 1]: (u + 1) * 2^-32 — never 0, safe for log()
ES_HD float u01_open(uint32_t u) {
  return ((float)u + 1.0f) * 2.3283064365386963e-10f;
}

// uniform in [0, 1)
ES_HD float u01(uint32_t u) {
  return (float)u * 2.3283064365386963e-10f;
}

struct f32x4 { float x, y, z, w; };

// 4 standard normals for "group" g under (seed, stream) via Box-Muller.
// Element i of a stream's normal sequence = component (i & 3) of group (i >> 2).
ES_HD f32x4 normal4(uint64_t g, uint64_t seed, uint32_t stream) {
  u32x4 r = philox4x32((uint32_t)g, (uint32_t)(g >> 32), stream, 0x6573616Du,  // "mase"
                       (uint32_t)seed, (uint32_t)(seed >> 32));
  const float TWO_PI = 6.2831853071795864769f;
  float u1 = u01_open(r.x), u2 = u01(r.y);
  float u3 = u01_open(r.z), u4 = u01(r.w);
  float r1 = sqrtf(-2.0f * logf(u1));
  float r2 = sqrtf(-2.0f * logf(u3));
  f32x4 out;
  out.x = r1 * cosf(TWO_PI * u2);
  out.y = r1 * sinf(TWO_PI * u2);
  out.z = r2 * cosf(TWO_PI * u4);
  out.w = r2 * sinf(TWO_PI * u4);
  return out;
}

}  // namespace esrng
