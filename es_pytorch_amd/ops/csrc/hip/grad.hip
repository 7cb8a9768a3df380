// K5 — gather-GEMV gradient reconstruction (the flagship op).
//
//   g[t] = sum_p fits[p] * table[offsets[p] + t]        t in [0, n_params)
//
// The reference does this as batched numpy dots over copied noise rows
// (src/utils/utils.py:29-39, batch_size=500 rows). Here the rows are never
// copied: each thread owns ELEMS consecutive output elements (strided across
// the block for coalescing) and walks the whole population, reading each
// row's slice directly from the HBM table. (fits, offset) pairs are staged
// through LDS in tiles. Memory-bound: pop x n_params x 4B table reads.
#include "common.h"

#define ES_GRAD_TILE 256

// Q: re-quantize each table value through the e4m3 round trip at sigma
// `qstd` before the FMA — the fp8-eps rollout evaluated fitness at
// perturbations q(sigma*eps)/sigma, and an ES estimator is exact when the
// gather uses the SAME perturbation values it evaluated (g = sum f_p * e'_p).
template <int ES_GRAD_ELEMS, bool Q>
__global__ void __launch_bounds__(256)
grad_gather_kernel(float* __restrict__ g, const float* __restrict__ table,
                   const float* __restrict__ fits, const int64_t* __restrict__ offsets,
                   int64_t n_pop, int64_t n_params, float qstd) {
  const float qinv = Q ? 1.0f / qstd : 0.0f;
  auto xform = [&](float x) { return Q ? e4m3_roundtrip(qstd * x) * qinv : x; };
  __shared__ float s_fit[ES_GRAD_TILE];
  __shared__ int64_t s_off[ES_GRAD_TILE];

  const int64_t chunk = (int64_t)blockDim.x * ES_GRAD_ELEMS;
  const int64_t base = blockIdx.x * chunk;
  float acc[ES_GRAD_ELEMS];
#pragma unroll
  for (int k = 0; k < ES_GRAD_ELEMS; ++k) acc[k] = 0.0f;

  for (int64_t p0 = 0; p0 < n_pop; p0 += ES_GRAD_TILE) {
    const int tile = (int)min((int64_t)ES_GRAD_TILE, n_pop - p0);
    __syncthreads();
    for (int i = threadIdx.x; i < tile; i += blockDim.x) {
      s_fit[i] = fits[p0 + i];
      s_off[i] = offsets[p0 + i];
    }
    __syncthreads();
    // software-pipelined over population rows: row i+1's 4 loads issue
    // before row i's FMAs (hipcc alone drains vmcnt per row — latency-bound,
    // measured ~7x off the bandwidth bound)
    const bool full = base + chunk <= n_params;
    if (full && tile > 1) {
      float c[ES_GRAD_ELEMS];
      const float* row0 = table + s_off[0] + base;
#pragma unroll
      for (int k = 0; k < ES_GRAD_ELEMS; ++k)
        c[k] = xform(row0[k * blockDim.x + threadIdx.x]);
      for (int i = 0; i < tile - 1; ++i) {
        const float* rown = table + s_off[i + 1] + base;
        float nx[ES_GRAD_ELEMS];
#pragma unroll
        for (int k = 0; k < ES_GRAD_ELEMS; ++k)
          nx[k] = xform(rown[k * blockDim.x + threadIdx.x]);
        const float f = s_fit[i];
#pragma unroll
        for (int k = 0; k < ES_GRAD_ELEMS; ++k) {
          acc[k] = fmaf(f, c[k], acc[k]);
          c[k] = nx[k];
        }
      }
      const float f = s_fit[tile - 1];
#pragma unroll
      for (int k = 0; k < ES_GRAD_ELEMS; ++k) acc[k] = fmaf(f, c[k], acc[k]);
    } else {
      for (int i = 0; i < tile; ++i) {
        const float f = s_fit[i];
        const float* row = table + s_off[i] + base;
#pragma unroll
        for (int k = 0; k < ES_GRAD_ELEMS; ++k) {
          const int64_t t = base + k * blockDim.x + threadIdx.x;
          if (t < n_params)
            acc[k] = fmaf(f, xform(row[k * blockDim.x + threadIdx.x]), acc[k]);
        }
      }
    }
  }

#pragma unroll
  for (int k = 0; k < ES_GRAD_ELEMS; ++k) {
    const int64_t t = base + k * blockDim.x + threadIdx.x;
    if (t < n_params) g[t] = acc[k];
  }
}

extern "C" int es_grad_gather(void* g, const void* table, const void* fits,
                              const void* offsets, int64_t n_pop, int64_t n_params,
                              float qstd, void* stream) {
  const int threads = 256;
  // pick the per-thread element count so the grid fills the chip (a fixed
  // ELEMS=4 gave only 163 blocks at n=167k — occupancy-starved)
  int elems = 4;
  while (elems > 1 && (n_params + (int64_t)threads * elems - 1) /
                          ((int64_t)threads * elems) < 1024)
    elems >>= 1;
  const int64_t chunk = (int64_t)threads * elems;
  const int blocks = (int)((n_params + chunk - 1) / chunk);
  const bool q = qstd > 0.0f;
#define ES_LAUNCH_GG(E, QF)                                                       \
  grad_gather_kernel<E, QF><<<dim3(blocks), dim3(threads), 0,                     \
                              (hipStream_t)stream>>>(                             \
      (float*)g, (const float*)table, (const float*)fits,                         \
      (const int64_t*)offsets, n_pop, n_params, qstd)
  if (elems == 4) { if (q) ES_LAUNCH_GG(4, true); else ES_LAUNCH_GG(4, false); }
  else if (elems == 2) { if (q) ES_LAUNCH_GG(2, true); else ES_LAUNCH_GG(2, false); }
  else { if (q) ES_LAUNCH_GG(1, true); else ES_LAUNCH_GG(1, false); }
#undef ES_LAUNCH_GG
  ES_CHECK_LAUNCH();
  return 0;
}
