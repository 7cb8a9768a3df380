// K3 — standalone population-batched MLP policy forward.
//
// One workgroup per population member evaluates the member's ENTIRE MLP:
// observation normalization (clamp((ob-mean)/std, +-clip), reference
// src/nn/nn.py:45), every Linear+Tanh layer (core scheme in mlp_core.h) and
// gaussian action noise (reference nn.py:47-48). Used by the generic-env
// engine path; the locomotion bench path uses the fully fused
// rollout_loco.hip instead.
#include "mlp_core.h"

__global__ void __launch_bounds__(256)
mlp_fwd_kernel(float* __restrict__ actions, const float* __restrict__ obs,
               const uint16_t* __restrict__ weights, const float* __restrict__ obmean,
               const float* __restrict__ obstd, MlpShape sh, float ob_clip,
               const float* __restrict__ ac_std_dev,
               const uint64_t* __restrict__ seed_dev, uint64_t salt, int64_t row_stride,
               int act_final, int noiseless_from, int bins, int eps, int act_mode,
               const float* __restrict__ alow, const float* __restrict__ arange) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* bufA = reinterpret_cast<float*>(smem);
  float* bufB = bufA + sh.maxdim;
  float* partial = bufB + sh.maxdim;
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const uint16_t* wb = weights + (int64_t)(b / eps) * row_stride;
  const int D = sh.dims[0];

  for (int i = tid; i < D; i += blockDim.x) {
    float v = (obs[(int64_t)b * D + i] - obmean[i]) / obstd[i];
    bufA[i] = fclampf(v, -ob_clip, ob_clip);
  }
  __syncthreads();

  const float* out = mlp_layers(wb, sh, bufA, bufB, partial, tid, blockDim.x, act_final);

  const int A = sh.dims[sh.n_layers];
  const uint64_t seed = seed_dev ? (*seed_dev + salt) : salt;
  if (act_mode == 2 || act_mode == 3) {
    // integrated gaussian actions (reference FFIntegGausAction(Multi),
    // nn.py:53-96): the net emits its own action std — first output (mode 2)
    // or the second half of the outputs (mode 3, abs)
    const int adim = act_mode == 2 ? A - 1 : A / 2;
    for (int d = tid; d < adim; d += blockDim.x) {
      float a = act_mode == 2 ? out[1 + d] : out[d];
      const float std_o = act_mode == 2 ? out[0] : fabsf(out[adim + d]);
      if (b < noiseless_from && std_o != 0.0f)
        a += std_o * es_actnoise(seed, (uint64_t)b * A + d);
      actions[(int64_t)b * adim + d] = a;
    }
    return;
  }
  if (bins > 1) {
    // K9 binned-action decode (reference FFBinned, nn.py:111-117): per-dim
    // argmax over bins, rescaled into [alow, alow+range]
    const int adim = A / bins;
    for (int d = tid; d < adim; d += blockDim.x) {
      int best = 0;
      float bv = out[d * bins];
      for (int j = 1; j < bins; ++j) {
        const float v = out[d * bins + j];
        if (v > bv) { bv = v; best = j; }
      }
      actions[(int64_t)b * adim + d] =
          alow[d] + arange[d] * (float)best / (float)(bins - 1);
    }
    return;
  }
  // ac_std is read from device memory so decay schedules keep working under
  // hipGraph replay (kernel args are frozen at capture time)
  const float ac_std = ac_std_dev ? *ac_std_dev : 0.0f;
  for (int o = tid; o < A; o += blockDim.x) {
    float a = out[o];
    // members >= noiseless_from are evaluated without action noise
    // (the reference's noiseless eval passes rs=None, es.py:48)
    if (ac_std != 0.0f && b < noiseless_from)
      a += ac_std * es_actnoise(seed, (uint64_t)b * A + o);
    actions[(int64_t)b * A + o] = a;
  }
}

extern "C" int es_mlp_fwd(void* actions, const void* obs, const void* weights,
                          const void* obmean, const void* obstd, const int32_t* dims_host,
                          int32_t ndims, const void* seed_dev, uint64_t salt, int32_t n_pop,
                          float ob_clip, const void* ac_std_dev, int64_t row_stride,
                          int32_t act_final, int32_t noiseless_from, int32_t bins,
                          int32_t eps, int32_t act_mode, const void* alow,
                          const void* arange, void* stream) {
  MlpShape sh;
  int rc = mlp_shape_init(&sh, dims_host, ndims, row_stride);
  if (rc) return rc;
  mlp_fwd_kernel<<<dim3((unsigned)n_pop), dim3(256), (unsigned)mlp_lds_bytes(sh.maxdim),
                   (hipStream_t)stream>>>(
      (float*)actions, (const float*)obs, (const uint16_t*)weights, (const float*)obmean,
      (const float*)obstd, sh, ob_clip, (const float*)ac_std_dev,
      (const uint64_t*)seed_dev, salt, row_stride, act_final, noiseless_from, bins,
      eps > 0 ? eps : 1, act_mode, (const float*)alow, (const float*)arange);
  ES_CHECK_LAUNCH();
  return 0;
}
