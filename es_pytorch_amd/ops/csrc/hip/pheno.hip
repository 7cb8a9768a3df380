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
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  // 4-deep pipeline: keep 8 loads in flight (scalar loads — the table slice
  // has arbitrary 4 B alignment, so float4 is not available)
  for (; t + 3 * stride < n_params; t += 4 * stride) {
    const float th0 = theta[t], n0 = noise[t];
    const float th1 = theta[t + stride], n1 = noise[t + stride];
    const float th2 = theta[t + 2 * stride], n2 = noise[t + 2 * stride];
    const float th3 = theta[t + 3 * stride], n3 = noise[t + 3 * stride];
    ob[t] = f2bf(th0 + s * n0);
    ob[t + stride] = f2bf(th1 + s * n1);
    ob[t + 2 * stride] = f2bf(th2 + s * n2);
    ob[t + 3 * stride] = f2bf(th3 + s * n3);
  }
  for (; t < n_params; t += stride) ob[t] = f2bf(theta[t] + s * noise[t]);
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
