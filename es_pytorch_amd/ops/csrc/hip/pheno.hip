// K2 — batched antithetic perturbation ("pheno") kernel.
//
// Materializes the whole population's perturbed parameter sets in one launch:
//   out[b][t] = bf16( theta[t] + sign[b] * std * table[offset[b] + t] )
// vs the reference's one-at-a-time host pheno (src/core/policy.py:61-67).
//
// `theta` is stored in the engine's FORWARD layout (per layer: W^T then b) so
// every access here and in mlp_fwd/grad is coalesced; the flat<->forward
// permutation lives host-side in the engine. sign[b] = +1/-1 for antithetic
// pairs and 0 for the noiseless-evaluation slot (then out = theta exactly).
#include "common.h"

__global__ void pheno_bf16_kernel(uint16_t* __restrict__ out, const float* __restrict__ theta,
                                  const float* __restrict__ table,
                                  const int64_t* __restrict__ offsets,
                                  const float* __restrict__ signs, int64_t n_params,
                                  int64_t row_stride, float std) {
  int64_t b = blockIdx.y;
  const float s = signs[b] * std;
  const float* noise = table + offsets[b];
  uint16_t* ob = out + b * row_stride;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n_params;
       t += (int64_t)gridDim.x * blockDim.x) {
    ob[t] = f2bf(theta[t] + s * noise[t]);
  }
}

extern "C" int es_pheno_bf16(void* out, const void* theta, const void* table,
                             const void* offsets, const void* signs, int64_t n_pop,
                             int64_t n_params, int64_t row_stride, float std, void* stream) {
  int threads = 256;
  int bx = (int)std::min<int64_t>((n_params + threads - 1) / threads, 1024);
  dim3 grid(bx, (unsigned)n_pop);
  pheno_bf16_kernel<<<grid, dim3(threads), 0, (hipStream_t)stream>>>(
      (uint16_t*)out, (const float*)theta, (const float*)table, (const int64_t*)offsets,
      (const float*)signs, n_params, row_stride, std);
  ES_CHECK_LAUNCH();
  return 0;
}
