// K6 — fused optimizer step on the flat parameter vector.
//
// Implements in ONE elementwise pass what the reference composes from numpy
// temporaries: the gradient scaling g/n_ranked (src/core/es.py:100), the
// combined input l2coeff*theta - grad (es.py:101), the Adam moment updates
// with bias correction (src/nn/optimizers.py:53-61) or SGD momentum
// (optimizers.py:39-44), and the parameter add (src/core/policy.py:73-74).
// Sign convention preserved: Adam/SGD return the NEGATIVE step of their
// input, so theta += -step(l2*theta - g) == ascent on fitness with L2 decay.
//
// theta/m/v are fp32 in the engine's forward layout; `a` (the bias-corrected
// lr) is computed host-side from t like the reference.
#include "common.h"

__global__ void adam_step_kernel(float* __restrict__ theta, float* __restrict__ m,
                                 float* __restrict__ v, const float* __restrict__ g,
                                 int64_t n, float a, float b1, float b2, float eps, float l2,
                                 float gscale) {
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n;
       t += (int64_t)gridDim.x * blockDim.x) {
    const float th = theta[t];
    const float gg = l2 * th - g[t] * gscale;
    const float mn = b1 * m[t] + (1.0f - b1) * gg;
    const float vn = b2 * v[t] + (1.0f - b2) * gg * gg;
    m[t] = mn;
    v[t] = vn;
    theta[t] = th - a * mn / (sqrtf(vn) + eps);
  }
}

__global__ void sgd_step_kernel(float* __restrict__ theta, float* __restrict__ v,
                                const float* __restrict__ g, int64_t n, float lr,
                                float momentum, float l2, float gscale) {
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n;
       t += (int64_t)gridDim.x * blockDim.x) {
    const float th = theta[t];
    const float gg = l2 * th - g[t] * gscale;
    const float vn = momentum * v[t] + (1.0f - momentum) * gg;
    v[t] = vn;
    theta[t] = th - lr * vn;
  }
}

extern "C" int es_adam_step(void* theta, void* m, void* v, const void* g, int64_t n, float a,
                            float b1, float b2, float eps, float l2, float gscale,
                            void* stream) {
  int threads = 256;
  int blocks = (int)std::min<int64_t>((n + threads - 1) / threads, 2048);
  adam_step_kernel<<<dim3(blocks), dim3(threads), 0, (hipStream_t)stream>>>(
      (float*)theta, (float*)m, (float*)v, (const float*)g, n, a, b1, b2, eps, l2, gscale);
  ES_CHECK_LAUNCH();
  return 0;
}

extern "C" int es_sgd_step(void* theta, void* v, const void* g, int64_t n, float lr,
                           float momentum, float l2, float gscale, void* stream) {
  int threads = 256;
  int blocks = (int)std::min<int64_t>((n + threads - 1) / threads, 2048);
  sgd_step_kernel<<<dim3(blocks), dim3(threads), 0, (hipStream_t)stream>>>(
      (float*)theta, (float*)v, (const float*)g, n, lr, momentum, l2, gscale);
  ES_CHECK_LAUNCH();
  return 0;
}
