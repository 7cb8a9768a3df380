// CPU-side native ops: deterministic Philox noise-table fill and the
// fitness-weighted noise sum (gradient reconstruction), multi-threaded.
//
// The reference does both in numpy on the host (noise fill:
// src/core/noisetable.py:61-64; gradient: src/utils/utils.py:29-39). These
// C++ versions are the host-side twins of the HIP kernels in hip/ — used by
// CPU tests and the CPU episodic path; the GPU engine uses the HIP versions.
//
// Exposed as a plain C ABI and loaded via ctypes (no torch ABI dependency):
//   g++ -O3 -march=native -shared -fPIC cpu_ops.cpp -o _cpu_ops.so
#include <stdint.h>
#include <string.h>

#include <algorithm>
#include <thread>
#include <vector>

#include "philox.h"

namespace {

void fill_range(float* out, int64_t begin, int64_t end, uint64_t seed, uint32_t stream) {
  // begin/end are element indices; groups of 4 share one philox call
  int64_t i = begin;
  while (i < end) {
    uint64_t g = (uint64_t)i >> 2;
    esrng::f32x4 n = esrng::normal4(g, seed, stream);
    const float v[4] = {n.x, n.y, n.z, n.w};
    int64_t gbase = (int64_t)(g << 2);
    for (int k = (int)(i - gbase); k < 4 && gbase + k < end; ++k) {
      out[gbase + k] = v[k];
      ++i;
    }
  }
}

int nthreads_for(int64_t n) {
  unsigned hw = std::thread::hardware_concurrency();
  if (hw == 0) hw = 4;
  int64_t per = 1 << 20;
  return (int)std::max<int64_t>(1, std::min<int64_t>(hw, (n + per - 1) / per));
}

}  // namespace

extern "C" {

// out[i] = N(0,1) from philox(seed, stream) at absolute index i
void es_noise_fill_cpu(float* out, int64_t n, uint64_t seed, uint32_t stream) {
  int nt = nthreads_for(n);
  if (nt == 1) {
    fill_range(out, 0, n, seed, stream);
    return;
  }
  std::vector<std::thread> ts;
  int64_t chunk = (n + nt - 1) / nt;
  chunk = (chunk + 3) & ~3;  // group-aligned so threads never split a philox group
  for (int t = 0; t < nt; ++t) {
    int64_t b = t * chunk, e = std::min<int64_t>(n, b + chunk);
    if (b >= e) break;
    ts.emplace_back(fill_range, out, b, e, seed, stream);
  }
  for (auto& t : ts) t.join();
}

// g[i] = sum_p fits[p] * table[offsets[p] + i]   (i in [0, n_params))
// The CPU twin of hip/grad.hip's gather-GEMV (reference utils.py:29-39).
void es_grad_gather_cpu(float* g, const float* table, const float* fits,
                        const int64_t* offsets, int64_t n_pop, int64_t n_params) {
  int nt = nthreads_for(n_params * std::max<int64_t>(1, n_pop / 8));
  auto work = [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) g[i] = 0.0f;
    for (int64_t p = 0; p < n_pop; ++p) {
      const float f = fits[p];
      if (f == 0.0f) continue;
      const float* row = table + offsets[p];
      for (int64_t i = b; i < e; ++i) g[i] += f * row[i];
    }
  };
  if (nt == 1) {
    work(0, n_params);
    return;
  }
  std::vector<std::thread> ts;
  int64_t chunk = (n_params + nt - 1) / nt;
  for (int t = 0; t < nt; ++t) {
    int64_t b = t * chunk, e = std::min<int64_t>(n_params, b + chunk);
    if (b >= e) break;
    ts.emplace_back(work, b, e);
  }
  for (auto& t : ts) t.join();
}

}  // extern "C"
