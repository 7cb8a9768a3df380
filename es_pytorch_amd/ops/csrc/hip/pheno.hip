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
  // thread owns 8 consecutive elements: theta reads as 2x float4 and the
  // bf16 row (16 B-aligned via row_stride padding) writes as ONE 16 B store;
  // the noise slice has arbitrary 4 B alignment, so its 8 loads stay scalar
  // (independent -> pipelined). 2 B/lane stores were the bottleneck before.
  typedef uint16_t u16x8 __attribute__((ext_vector_type(8)));
  const int64_t stride8 = (int64_t)gridDim.x * blockDim.x * 8;
  int64_t t = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  for (; t + 7 < n_params; t += stride8) {
    const float4 th0 = *reinterpret_cast<const float4*>(theta + t);
    const float4 th1 = *reinterpret_cast<const float4*>(theta + t + 4);
    float nz[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) nz[k] = noise[t + k];
    u16x8 v;
    v[0] = f2bf(th0.x + s * nz[0]);
    v[1] = f2bf(th0.y + s * nz[1]);
    v[2] = f2bf(th0.z + s * nz[2]);
    v[3] = f2bf(th0.w + s * nz[3]);
    v[4] = f2bf(th1.x + s * nz[4]);
    v[5] = f2bf(th1.y + s * nz[5]);
    v[6] = f2bf(th1.z + s * nz[6]);
    v[7] = f2bf(th1.w + s * nz[7]);
    *reinterpret_cast<u16x8*>(ob + t) = v;
  }
  for (; t < n_params; ++t) ob[t] = f2bf(theta[t] + s * noise[t]);
}

extern "C" int es_pheno_bf16(void* out, const void* theta, const void* table,
                             const void* offsets, const void* signs, int64_t n_pop,
                             int64_t n_params, int64_t row_stride, float std, void* stream) {
  int threads = 256;
  int bx = (int)std::min<int64_t>((n_params + threads * 8 - 1) / (threads * 8), 1024);
  dim3 grid(bx, (unsigned)n_pop);
  pheno_bf16_kernel<<<grid, dim3(threads), 0, (hipStream_t)stream>>>(
      (uint16_t*)out, (const float*)theta, (const float*)table, (const int64_t*)offsets,
      (const float*)signs, n_params, row_stride, std);
  ES_CHECK_LAUNCH();
  return 0;
}
