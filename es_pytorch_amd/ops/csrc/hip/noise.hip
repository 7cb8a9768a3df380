// K1 — HBM-resident noise-table fill.
//
// Replaces the reference's one-rank-per-node numpy randn fill of an MPI
// shared window (reference src/core/noisetable.py:61-64,85-88): every GPU
// fills its own table in HBM deterministically from a broadcast 64-bit seed.
// Grid-stride over Philox groups of 4 elements; bitwise deterministic for a
// given (seed, stream) regardless of grid shape.
#include "common.h"

__global__ void noise_fill_kernel(float* __restrict__ out, int64_t n, uint64_t seed,
                                  uint32_t stream_id) {
  int64_t ngroups = (n + 3) >> 2;
  for (int64_t g = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; g < ngroups;
       g += (int64_t)gridDim.x * blockDim.x) {
    esrng::f32x4 v = esrng::normal4((uint64_t)g, seed, stream_id);
    int64_t base = g << 2;
    if (base + 4 <= n) {
      *reinterpret_cast<float4*>(out + base) = make_float4(v.x, v.y, v.z, v.w);
    } else {
      const float vv[4] = {v.x, v.y, v.z, v.w};
      for (int k = 0; base + k < n; ++k) out[base + k] = vv[k];
    }
  }
}

extern "C" int es_noise_fill(void* out, int64_t n, uint64_t seed, uint32_t stream_id,
                             void* stream) {
  int threads = 256;
  int64_t ngroups = (n + 3) >> 2;
  int blocks = (int)std::min<int64_t>((ngroups + threads - 1) / threads, 8192);
  noise_fill_kernel<<<dim3(blocks), dim3(threads), 0, (hipStream_t)stream>>>(
      (float*)out, n, seed, stream_id);
  ES_CHECK_LAUNCH();
  return 0;
}
