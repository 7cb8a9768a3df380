// K3 — fused population-batched MLP policy forward.
//
// One workgroup per population member per env step evaluates the member's
// ENTIRE MLP: observation normalization (clamp((ob-mean)/std, +-clip),
// reference src/nn/nn.py:45), every Linear+Tanh layer, and gaussian action
// noise (reference nn.py:47-48) — all in one launch, activations staged in
// LDS, weights streamed from the member's bf16 blob in HBM.
//
// Layout contract (shared with pheno.hip / grad.hip / engine.py): per layer,
// W^T stored (in_dim, out_dim) then bias (out_dim). Thread t computes output
// pair o = 2t (ushort2 loads, coalesced across the wave) when the layer's
// offsets/dims allow, else scalar. The reference applies the activation after
// EVERY layer including the last (nn.py:36-37); act_final mirrors that.
//
// The workload is HBM-bandwidth-bound (population x n_params bf16 weight
// reads per step); this kernel's job is to stream weights at full coalescing
// with zero intermediate HBM traffic for activations.
#include "common.h"

#define ES_MAXL 8
#define ES_MAXDIM 2048

struct MlpShape {
  int n_layers;              // number of Linear layers
  int dims[ES_MAXL + 1];     // [in, h1, ..., out]
  int64_t woff[ES_MAXL];     // element offset of layer's W^T within a member blob
  int64_t boff[ES_MAXL];     // element offset of layer's bias
};

__device__ __forceinline__ float es_actnoise(uint64_t seed, uint64_t ctr) {
  esrng::f32x4 v = esrng::normal4(ctr, seed, 0xACu);
  return v.x;
}

__global__ void __launch_bounds__(256)
mlp_fwd_kernel(float* __restrict__ actions, const float* __restrict__ obs,
               const uint16_t* __restrict__ weights, const float* __restrict__ obmean,
               const float* __restrict__ obstd, MlpShape sh, float ob_clip, float ac_std,
               const uint64_t* __restrict__ seed_dev, uint64_t salt, int64_t n_params,
               int act_final) {
  __shared__ float buf[2][ES_MAXDIM];
  const int b = blockIdx.x;
  const uint16_t* wb = weights + (int64_t)b * n_params;
  const int D = sh.dims[0];

  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float v = (obs[(int64_t)b * D + i] - obmean[i]) / obstd[i];
    buf[0][i] = fclampf(v, -ob_clip, ob_clip);
  }
  __syncthreads();

  int cur = 0;
  for (int l = 0; l < sh.n_layers; ++l) {
    const int I = sh.dims[l], O = sh.dims[l + 1];
    const uint16_t* Wt = wb + sh.woff[l];
    const uint16_t* Bs = wb + sh.boff[l];
    const bool do_act = (l < sh.n_layers - 1) || act_final;
    const float* x = buf[cur];
    float* y = buf[cur ^ 1];

    if ((O & 1) == 0 && ((sh.woff[l] | sh.boff[l]) & 1) == 0) {
      // paired path: thread handles outputs (2t, 2t+1); wave reads 4 B/lane
      for (int op = threadIdx.x; op < (O >> 1); op += blockDim.x) {
        const int o = op << 1;
        float acc0 = bf2f(Bs[o]), acc1 = bf2f(Bs[o + 1]);
        const uint16_t* wrow = Wt + o;
        for (int i = 0; i < I; ++i) {
          const uint32_t w2 = *reinterpret_cast<const uint32_t*>(wrow + (int64_t)i * O);
          const float xi = x[i];
          acc0 = fmaf(bf2f((uint16_t)(w2 & 0xFFFFu)), xi, acc0);
          acc1 = fmaf(bf2f((uint16_t)(w2 >> 16)), xi, acc1);
        }
        y[o] = do_act ? tanhf(acc0) : acc0;
        y[o + 1] = do_act ? tanhf(acc1) : acc1;
      }
    } else {
      for (int o = threadIdx.x; o < O; o += blockDim.x) {
        float acc = bf2f(Bs[o]);
        for (int i = 0; i < I; ++i) acc = fmaf(bf2f(Wt[(int64_t)i * O + o]), x[i], acc);
        y[o] = do_act ? tanhf(acc) : acc;
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  const int A = sh.dims[sh.n_layers];
  const uint64_t seed = seed_dev ? (*seed_dev + salt) : salt;
  for (int o = threadIdx.x; o < A; o += blockDim.x) {
    float a = buf[cur][o];
    if (ac_std != 0.0f) a += ac_std * es_actnoise(seed, (uint64_t)b * A + o);
    actions[(int64_t)b * A + o] = a;
  }
}

extern "C" int es_mlp_fwd(void* actions, const void* obs, const void* weights,
                          const void* obmean, const void* obstd, const int32_t* dims_host,
                          int32_t ndims, const void* seed_dev, uint64_t salt, int32_t n_pop,
                          float ob_clip, float ac_std, int64_t n_params, int32_t act_final,
                          void* stream) {
  if (ndims < 2 || ndims > ES_MAXL + 1) return -100;
  MlpShape sh;
  sh.n_layers = ndims - 1;
  int64_t off = 0;
  for (int l = 0; l < ndims; ++l) {
    sh.dims[l] = dims_host[l];
    if (dims_host[l] > ES_MAXDIM) return -101;
  }
  for (int l = 0; l < sh.n_layers; ++l) {
    sh.woff[l] = off;
    off += (int64_t)sh.dims[l] * sh.dims[l + 1];
    sh.boff[l] = off;
    off += sh.dims[l + 1];
  }
  if (off != n_params) return -102;
  mlp_fwd_kernel<<<dim3((unsigned)n_pop), dim3(256), 0, (hipStream_t)stream>>>(
      (float*)actions, (const float*)obs, (const uint16_t*)weights, (const float*)obmean,
      (const float*)obstd, sh, ob_clip, ac_std, (const uint64_t*)seed_dev, salt, n_params,
      act_final);
  ES_CHECK_LAUNCH();
  return 0;
}
