// Fused rollout step for the synthetic locomotion envs: ONE kernel per env
// step does, per population member (one workgroup each):
//
//   policy MLP forward (mlp_core.h scheme, bf16 weight streaming)
//   + gaussian action noise (skipped for the noiseless slot, reference
//     es.py:48 evaluates noiselessly with rs=None)
//   + env dynamics  s' = (1-leak) s + leak tanh(s A + a B + b0)
//   + reward / positions / fall-termination (envs/locomotion.py semantics)
//   + alive-masked bookkeeping: total reward, per-member steps, behaviour
//     freeze, per-member observation sums for ObStat
//
// replacing ~25 small torch kernels + 1 forward launch per step (measured
// ~470 us/step) with a single HBM-bandwidth-bound launch. The A matrix is
// shared by all members (L2-resident); activations and state never leave
// LDS; actions never touch HBM.
#include "mlp_core.h"

struct LocoArgs {
  int S;            // latent state dim
  int A;            // action dim
  int D;            // obs dim (= S, or S+2 goal-conditioned)
  int goal;         // goal-conditioned flag
  int terminate;    // terminate_on_fall
  int noiseless_from;  // members >= this index get no action noise
  int bins;            // >1: K9 binned-action decode (FFBinned)
  float leak, ctrl, alive_bonus, fall_thr, dt, ob_clip;
  uint64_t salt;
  int64_t row_stride;
};

__global__ void __launch_bounds__(256)
loco_step_kernel(const uint16_t* __restrict__ weights, const float* __restrict__ obmean,
                 const float* __restrict__ obstd, MlpShape sh, LocoArgs la,
                 const float* __restrict__ ac_std_dev,
                 const uint64_t* __restrict__ seed_dev,
                 float* __restrict__ s_glob, float* __restrict__ pos,
                 const float* __restrict__ goal, const float* __restrict__ Am,
                 const float* __restrict__ Bm, const float* __restrict__ b0,
                 const float* __restrict__ wv, const float* __restrict__ wa,
                 const float* __restrict__ wy, const float* __restrict__ wh,
                 float* __restrict__ alive, float* __restrict__ rew_total,
                 float* __restrict__ member_steps, float* __restrict__ behv,
                 float* __restrict__ mo_sum, float* __restrict__ mo_sumsq) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* bufA = reinterpret_cast<float*>(smem);
  float* bufB = bufA + sh.maxdim;
  float* partial = bufB + sh.maxdim;
  float* raws = partial + 256 * 8;
  float* abuf = raws + ((la.S + 3) & ~3);
  float* sc = abuf + 64;  // [4]: alive_old broadcast
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  const int S = la.S, A = la.A;
  float* sb = s_glob + (int64_t)b * S;

  // ---- build normalized obs in buf[0]; keep raw state in LDS -------------
  for (int i = tid; i < S; i += nth) {
    const float v = sb[i];
    raws[i] = v;
    bufA[i] = fclampf((v - obmean[i]) / obstd[i], -la.ob_clip, la.ob_clip);
  }
  if (la.goal && tid < 2) {
    const float rel = (goal[(int64_t)b * 2 + tid] - pos[(int64_t)b * 3 + tid]) * 0.1f;
    bufA[S + tid] = fclampf((rel - obmean[S + tid]) / obstd[S + tid], -la.ob_clip,
                            la.ob_clip);
  }
  __syncthreads();

  // ---- policy forward ----------------------------------------------------
  const uint16_t* wb = weights + (int64_t)b * la.row_stride;
  const float* aout = mlp_layers(wb, sh, bufA, bufB, partial, tid, nth, 1);
  const uint64_t seed = seed_dev ? (*seed_dev + la.salt) : la.salt;
  const float ac_std = ac_std_dev ? *ac_std_dev : 0.0f;  // device-read: graph-safe decay
  if (la.bins > 1) {
    // K9 binned decode (FFBinned): per-dim argmax over bins -> [-1, 1]
    if (tid < A) {
      const float* row = aout + tid * la.bins;
      int best = 0;
      float bv = row[0];
      for (int j = 1; j < la.bins; ++j)
        if (row[j] > bv) { bv = row[j]; best = j; }
      abuf[tid] = -1.0f + 2.0f * (float)best / (float)(la.bins - 1);
    }
  } else if (tid < A) {
    float a = aout[tid];
    if (ac_std != 0.0f && b < la.noiseless_from)
      a += ac_std * es_actnoise(seed, (uint64_t)b * A + tid);
    abuf[tid] = fclampf(a, -1.0f, 1.0f);  // env action clamp (locomotion.py)
  }
  __syncthreads();

  // ---- dynamics: pre = s A + a B + b0 ; s' = (1-leak) s + leak tanh(pre) --
  {
    const bool quad = (S % 4 == 0);
    if (quad) {
      const int OCT = S >> 2;
      const int PART = nth / OCT;
      const int oi = tid % OCT, ip = tid / OCT;
      float acc[4] = {0, 0, 0, 0};
      if (ip < PART) {
        const float* acol = Am + (oi << 2);
        // software-pipelined register double-buffer (see mlp_core.h)
        auto ld = [&](int i) {
          return *reinterpret_cast<const float4*>(acol + (int64_t)i * S);
        };
        auto f4 = [&](const float4 w, const float xi) {
          acc[0] = fmaf(w.x, xi, acc[0]);
          acc[1] = fmaf(w.y, xi, acc[1]);
          acc[2] = fmaf(w.z, xi, acc[2]);
          acc[3] = fmaf(w.w, xi, acc[3]);
        };
        int i = ip;
        const int step4 = PART * 4;
        if (i + 3 * PART < S) {
          float4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
          for (; i + 7 * PART < S; i += step4) {
            const float4 n0 = ld(i + 4 * PART), n1 = ld(i + 5 * PART),
                         n2 = ld(i + 6 * PART), n3 = ld(i + 7 * PART);
            f4(c0, raws[i]);
            f4(c1, raws[i + PART]);
            f4(c2, raws[i + 2 * PART]);
            f4(c3, raws[i + 3 * PART]);
            c0 = n0; c1 = n1; c2 = n2; c3 = n3;
          }
          f4(c0, raws[i]);
          f4(c1, raws[i + PART]);
          f4(c2, raws[i + 2 * PART]);
          f4(c3, raws[i + 3 * PART]);
          i += step4;
        }
        for (; i < S; i += PART) f4(ld(i), raws[i]);
#pragma unroll
        for (int q = 0; q < 4; ++q) partial[(ip * OCT + oi) * 4 + q] = acc[q];
      }
      __syncthreads();
      for (int o = tid; o < S; o += nth) {
        float p = b0[o];
        const int oo = o >> 2, j = o & 3;
        for (int pp = 0; pp < PART; ++pp) p += partial[(pp * OCT + oo) * 4 + j];
        for (int k = 0; k < A; ++k) p = fmaf(abuf[k], Bm[(int64_t)k * S + o], p);
        const float sn = (1.0f - la.leak) * raws[o] + la.leak * tanhf(p);
        bufA[o] = sn;
        sb[o] = sn;
      }
    } else {
      for (int o = tid; o < S; o += nth) {
        float p = b0[o];
        for (int i = 0; i < S; ++i) p = fmaf(raws[i], Am[(int64_t)i * S + o], p);
        for (int k = 0; k < A; ++k) p = fmaf(abuf[k], Bm[(int64_t)k * S + o], p);
        const float sn = (1.0f - la.leak) * raws[o] + la.leak * tanhf(p);
        bufA[o] = sn;
        sb[o] = sn;
      }
    }
  }
  __syncthreads();

  // ---- block reductions: vfwd, vy, h, sum(a^2) ---------------------------
  {
    float p0 = 0, p1 = 0, p2 = 0, p3 = 0;
    for (int i = tid; i < S; i += nth) {
      const float sn = bufA[i];
      p0 = fmaf(sn, wv[i], p0);
      p1 = fmaf(sn, wy[i], p1);
      p2 = fmaf(sn, wh[i], p2);
    }
    for (int j = tid; j < A; j += nth) {
      p0 = fmaf(0.5f * abuf[j], wa[j], p0);
      p3 = fmaf(abuf[j], abuf[j], p3);
    }
    partial[tid] = p0;
    partial[256 + tid] = p1;
    partial[512 + tid] = p2;
    partial[768 + tid] = p3;
    __syncthreads();
    for (int off = nth >> 1; off > 0; off >>= 1) {
      if (tid < off) {
        partial[tid] += partial[tid + off];
        partial[256 + tid] += partial[256 + tid + off];
        partial[512 + tid] += partial[512 + tid + off];
        partial[768 + tid] += partial[768 + tid + off];
      }
      __syncthreads();
    }
  }

  // ---- scalar bookkeeping (thread 0) -------------------------------------
  if (tid == 0) {
    const float vfwd = partial[0], vy = partial[256], h = partial[512], asq = partial[768];
    const float alive_old = alive[b];
    float rew;
    if (la.goal) {
      const float rx = goal[(int64_t)b * 2 + 0] - pos[(int64_t)b * 3 + 0];
      const float ry = goal[(int64_t)b * 2 + 1] - pos[(int64_t)b * 3 + 1];
      const float inv = 1.0f / (sqrtf(rx * rx + ry * ry) + 1e-6f);
      rew = vfwd * rx * inv + vy * ry * inv - la.ctrl * asq + la.alive_bonus;
    } else {
      rew = vfwd - la.ctrl * asq + la.alive_bonus;
    }
    const float px = pos[(int64_t)b * 3 + 0] + la.dt * vfwd;
    const float py = pos[(int64_t)b * 3 + 1] + la.dt * vy;
    pos[(int64_t)b * 3 + 0] = px;
    pos[(int64_t)b * 3 + 1] = py;
    pos[(int64_t)b * 3 + 2] = h;
    const float done = (la.terminate && h < la.fall_thr) ? 1.0f : 0.0f;
    rew_total[b] += rew * alive_old;
    member_steps[b] += alive_old;
    if (alive_old > 0.0f) {
      behv[(int64_t)b * 3 + 0] = px;
      behv[(int64_t)b * 3 + 1] = py;
      behv[(int64_t)b * 3 + 2] = h;
    }
    alive[b] = alive_old * (1.0f - done);
    sc[4] = alive_old;
  }
  __syncthreads();

  // ---- per-member obs statistics (post-step obs, alive-weighted) ---------
  const float w = sc[4];
  if (w > 0.0f) {
    float* ms = mo_sum + (int64_t)b * la.D;
    float* mq = mo_sumsq + (int64_t)b * la.D;
    for (int i = tid; i < S; i += nth) {
      const float o = bufA[i];
      ms[i] += o;
      mq[i] += o * o;
    }
    if (la.goal && tid < 2) {
      const float rel = (goal[(int64_t)b * 2 + tid] - pos[(int64_t)b * 3 + tid]) * 0.1f;
      ms[S + tid] += rel;
      mq[S + tid] += rel * rel;
    }
  }
}

extern "C" int es_loco_step(const void* weights, const void* obmean, const void* obstd,
                            const int32_t* dims_host, int32_t ndims, const void* seed_dev,
                            uint64_t salt, float ob_clip, const void* ac_std_dev,
                            int64_t row_stride,
                            void* s_glob, void* pos, const void* goal, const void* Am,
                            const void* Bm, const void* b0, const void* wv, const void* wa,
                            const void* wy, const void* wh, void* alive, void* rew_total,
                            void* member_steps, void* behv, void* mo_sum, void* mo_sumsq,
                            int32_t n_pop, int32_t sdim, int32_t adim, int32_t goal_flag,
                            int32_t terminate, int32_t noiseless_from, int32_t bins,
                            float leak, float ctrl, float alive_bonus, float fall_thr,
                            float dt, void* stream) {
  MlpShape sh;
  int rc = mlp_shape_init(&sh, dims_host, ndims, row_stride);
  if (rc) return rc;
  if (adim > 64 || sdim > ES_MAXDIM) return -103;
  if (sh.dims[0] != sdim + (goal_flag ? 2 : 0)) return -104;
  const int out_dim = bins > 1 ? adim * bins : adim;
  if (sh.dims[sh.n_layers] != out_dim) return -105;
  LocoArgs la;
  la.S = sdim; la.A = adim; la.D = sh.dims[0]; la.goal = goal_flag;
  la.terminate = terminate; la.noiseless_from = noiseless_from; la.bins = bins;
  la.leak = leak; la.ctrl = ctrl; la.alive_bonus = alive_bonus; la.fall_thr = fall_thr;
  la.dt = dt; la.ob_clip = ob_clip; la.salt = salt;
  la.row_stride = row_stride;
  const unsigned lds = (unsigned)(mlp_lds_bytes(sh.maxdim) +
                                  (((sdim + 3) & ~3) + 64 + 8) * 4);
  loco_step_kernel<<<dim3((unsigned)n_pop), dim3(256), lds, (hipStream_t)stream>>>(
      (const uint16_t*)weights, (const float*)obmean, (const float*)obstd, sh, la,
      (const float*)ac_std_dev, (const uint64_t*)seed_dev, (float*)s_glob, (float*)pos, (const float*)goal,
      (const float*)Am, (const float*)Bm, (const float*)b0, (const float*)wv,
      (const float*)wa, (const float*)wy, (const float*)wh, (float*)alive,
      (float*)rew_total, (float*)member_steps, (float*)behv, (float*)mo_sum,
      (float*)mo_sumsq);
  ES_CHECK_LAUNCH();
  return 0;
}
