// Fused rollout kernels for the synthetic locomotion envs.
//
// Per population member (one workgroup), one env step comprises:
//   policy MLP forward (mlp_core.h scheme, bf16 weight streaming)
//   + gaussian action noise or K9 binned decode (skipped / noise-free for
//     the noiseless evaluation, reference es.py:48 passes rs=None)
//   + env dynamics  s' = (1-leak) s + leak tanh(s A + a B + b0)
//   + reward / positions / fall-termination (envs/locomotion.py semantics)
//   + alive-masked bookkeeping: total reward, per-member steps, behaviour
//     freeze, per-member observation sums for ObStat
//
// Two launch shapes share the body:
//  * es_loco_step    — ONE step for a whole population (grid = pop). The
//    engine replays max_steps of these from a hipGraph. Replaced ~25 small
//    torch kernels + a forward launch per step (measured ~470 us/step).
//  * es_loco_episode — a WHOLE episode for a few members (internal step
//    loop). Used for the noiseless evaluation on a side stream so the main
//    per-step grid stays an exact multiple of the CU slot count (a +1
//    straggler block measured ~15-20% tail on every step).
#include "mlp_core.h"

// ES_DYN_DEPTH8: 8-deep lookahead on the shared-A L2 stream (12 uint4 live
// instead of 8). Measured -4.3% same-box on the fp8 flagship (66.9 ->
// 64.1-64.4 us/step); the fp8 kernel lands at 123 VGPRs (4 waves kept) and
// the bf16 pair kernel stays 168 (its peak is in the forward). Default ON;
// accumulation order (ascending i) unchanged -> bitwise-same results.
#ifndef ES_DYN_DEPTH8
#define ES_DYN_DEPTH8 1
#endif

struct LocoArgs {
  int S;               // latent state dim
  int A;               // action dim
  int D;               // obs dim (= S, or S+2 goal-conditioned)
  int goal;            // goal-conditioned flag
  int terminate;       // terminate_on_fall
  int noiseless_from;  // members >= this index get no action noise
  int bins;            // >1: K9 binned-action decode (FFBinned)
  int act_mode;        // 0 plain, 1 binned, 2 integ-gauss, 3 integ-gauss-multi
  int eps;             // episodes per perturbation: slot b uses weights row b/eps
  int wrow0;           // weights-row index of slot 0 (episode kernel on a
                       // sub-blob, e.g. the pair path's 1-row theta blob)
  float leak, ctrl, alive_bonus, fall_thr, dt, ob_clip;
  int64_t row_stride;
};

struct LocoPtrs {
  const uint16_t* weights;
  const float *obmean, *obstd, *ac_std_dev;
  const uint64_t* seed_dev;
  float *s_glob, *pos;
  const float *goal, *Bm, *b0, *wv, *wa, *wy, *wh;
  const uint16_t* Am;  // bf16 state-transition matrix (S, S)
  float *alive, *rew_total, *member_steps, *behv, *mo_sum, *mo_sumsq;
};

// ---- build normalized obs for slot b; keep raw state in LDS --------------
__device__ __forceinline__ void loco_build_obs(
    const LocoArgs& la, const LocoPtrs& P, int b, float* bufA, float* raws,
    int tid, int nth) {
  const int S = la.S;
  const float* sb = P.s_glob + (int64_t)b * S;
  for (int i = tid; i < S; i += nth) {
    const float v = sb[i];
    raws[i] = v;
    bufA[i] = fclampf((v - P.obmean[i]) / P.obstd[i], -la.ob_clip, la.ob_clip);
  }
  if (la.goal && tid < 2) {
    const float rel = (P.goal[(int64_t)b * 2 + tid] - P.pos[(int64_t)b * 3 + tid]) * 0.1f;
    bufA[S + tid] = fclampf((rel - P.obmean[S + tid]) / P.obstd[S + tid], -la.ob_clip,
                            la.ob_clip);
  }
}

// ---- net output -> clamped env action for slot b (modes 0/1/2/3) ---------
__device__ __forceinline__ void loco_decode_action(
    const MlpShape& sh, const LocoArgs& la, const LocoPtrs& P, int b, uint64_t salt,
    const float* aout, float* abuf, int tid) {
  const int A = la.A;
  const uint64_t seed = P.seed_dev ? (*P.seed_dev + salt) : salt;
  const float ac_std = P.ac_std_dev ? *P.ac_std_dev : 0.0f;
  if (la.act_mode == 2 || la.act_mode == 3) {
    // integrated gaussian actions (reference nn.py:53-96)
    const int odim = sh.dims[sh.n_layers];
    if (tid < A) {
      float a = la.act_mode == 2 ? aout[1 + tid] : aout[tid];
      const float std_o = la.act_mode == 2 ? aout[0] : fabsf(aout[A + tid]);
      if (b < la.noiseless_from && std_o != 0.0f)
        a += std_o * es_actnoise(seed, (uint64_t)b * odim + tid);
      abuf[tid] = fclampf(a, -1.0f, 1.0f);
    }
  } else if (la.bins > 1) {
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
}

__device__ __forceinline__ void loco_fwd_body(
    const MlpShape& sh, const LocoArgs& la, const LocoPtrs& P, int b, uint64_t salt,
    float* bufA, float* bufB, float* partial, float* raws, float* abuf) {
  const int tid = threadIdx.x;
  const int nth = blockDim.x;
  loco_build_obs(la, P, b, bufA, raws, tid, nth);
  __syncthreads();
  const uint16_t* wb = P.weights + (int64_t)(b / la.eps - la.wrow0) * la.row_stride;
  const float* aout = mlp_layers(wb, sh, bufA, bufB, partial, tid, nth, 1);
  loco_decode_action(sh, la, P, b, salt, aout, abuf, tid);
  __syncthreads();
}

// A-matvec partial sums for ONE member (bf16 A, octet-tiled like the policy
// layers in mlp_core.h; S % 8 == 0 required). Writes the (PART, OCT, 8)
// partial layout consumed by loco_dyn_finish's output sweep.
__device__ __forceinline__ void loco_dyn_partials(
    const uint16_t* Am, int S, const float* raws, float* partial, int tid, int nth) {
  const int OCT = S >> 3;
  const int PART = nth / OCT;
  const int oi = tid % OCT, ip = tid / OCT;
  float acc[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) acc[q] = 0.0f;
  if (ip < PART) {
    const uint16_t* acol = Am + (oi << 3);
    auto ld = [&](int i) {
      return *reinterpret_cast<const uint4*>(acol + (int64_t)i * S);
    };
    int i = ip;
    const int step4 = PART * 4;
#if ES_DYN_DEPTH8
    // 8-deep lookahead on the L2 A stream (see loco_dyn_partials_pair)
    if (i + 7 * PART < S) {
      uint4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
      uint4 d0 = ld(i + 4 * PART), d1 = ld(i + 5 * PART), d2 = ld(i + 6 * PART),
            d3 = ld(i + 7 * PART);
      for (; i + 11 * PART < S; i += step4) {
        const uint4 n0 = ld(i + 8 * PART), n1 = ld(i + 9 * PART),
                    n2 = ld(i + 10 * PART), n3 = ld(i + 11 * PART);
        bf8_fma(c0, raws[i], acc);
        bf8_fma(c1, raws[i + PART], acc);
        bf8_fma(c2, raws[i + 2 * PART], acc);
        bf8_fma(c3, raws[i + 3 * PART], acc);
        c0 = d0; c1 = d1; c2 = d2; c3 = d3;
        d0 = n0; d1 = n1; d2 = n2; d3 = n3;
      }
      if (i < S) bf8_fma(c0, raws[i], acc);
      if (i + PART < S) bf8_fma(c1, raws[i + PART], acc);
      if (i + 2 * PART < S) bf8_fma(c2, raws[i + 2 * PART], acc);
      if (i + 3 * PART < S) bf8_fma(c3, raws[i + 3 * PART], acc);
      if (i + 4 * PART < S) bf8_fma(d0, raws[i + 4 * PART], acc);
      if (i + 5 * PART < S) bf8_fma(d1, raws[i + 5 * PART], acc);
      if (i + 6 * PART < S) bf8_fma(d2, raws[i + 6 * PART], acc);
      if (i + 7 * PART < S) bf8_fma(d3, raws[i + 7 * PART], acc);
      i += 8 * PART;
    }
#else
    if (i + 3 * PART < S) {
      uint4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
      for (; i + 7 * PART < S; i += step4) {
        const uint4 n0 = ld(i + 4 * PART), n1 = ld(i + 5 * PART),
                    n2 = ld(i + 6 * PART), n3 = ld(i + 7 * PART);
        bf8_fma(c0, raws[i], acc);
        bf8_fma(c1, raws[i + PART], acc);
        bf8_fma(c2, raws[i + 2 * PART], acc);
        bf8_fma(c3, raws[i + 3 * PART], acc);
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      }
      bf8_fma(c0, raws[i], acc);
      bf8_fma(c1, raws[i + PART], acc);
      bf8_fma(c2, raws[i + 2 * PART], acc);
      bf8_fma(c3, raws[i + 3 * PART], acc);
      i += step4;
    }
#endif
    for (; i < S; i += PART) bf8_fma(ld(i), raws[i], acc);
#pragma unroll
    for (int q = 0; q < 8; ++q) partial[(ip * OCT + oi) * 8 + q] = acc[q];
  }
  __syncthreads();
}

// Dynamics output sweep + fused epilogue for ONE member. The sweep that
// materializes s' also accumulates the reward/behaviour reduction partials
// AND the per-member obs statistics in the same pass (one s'-sweep instead
// of three, and wave shuffles replace the 8-barrier LDS reduction tree).
// When S % 8 == 0 the A-matvec partials must already be in `partial`
// (loco_dyn_partials); otherwise the scalar fallback reads Am directly.
__device__ __forceinline__ void loco_dyn_finish(
    const LocoArgs& la, const LocoPtrs& P, int b, const float* raws,
    const float* abuf, float* partial, int tid, int nth) {
  const int S = la.S, A = la.A;
  float* sb = P.s_glob + (int64_t)b * S;
  const float w_alive = P.alive[b];  // pre-step alive, used as obstat weight
  float p0 = 0, p1 = 0, p2 = 0, p3 = 0;
  float* ms = P.mo_sum + (int64_t)b * la.D;
  float* mq = P.mo_sumsq + (int64_t)b * la.D;
  {
    const bool oct8 = (S % 8 == 0);
    const int OCT = S >> 3;
    const int PART = oct8 ? nth / OCT : 0;
    for (int o = tid; o < S; o += nth) {
      float p = P.b0[o];
      if (oct8) {
        const int oo = o >> 3, j = o & 7;
        for (int pp = 0; pp < PART; ++pp) p += partial[(pp * OCT + oo) * 8 + j];
      } else {
        for (int i = 0; i < S; ++i)
          p = fmaf(raws[i], bf2f(P.Am[(int64_t)i * S + o]), p);
      }
      for (int k = 0; k < A; ++k) p = fmaf(abuf[k], P.Bm[(int64_t)k * S + o], p);
      const float sn = (1.0f - la.leak) * raws[o] + la.leak * tanhf(p);
      sb[o] = sn;
      p0 = fmaf(sn, P.wv[o], p0);
      p1 = fmaf(sn, P.wy[o], p1);
      p2 = fmaf(sn, P.wh[o], p2);
      if (w_alive > 0.0f) {
        ms[o] += sn;
        mq[o] += sn * sn;
      }
    }
    for (int j = tid; j < A; j += nth) {
      p0 = fmaf(0.5f * abuf[j], P.wa[j], p0);
      p3 = fmaf(abuf[j], abuf[j], p3);
    }
  }
  // wave-level shuffle reduction, then one cross-wave pass through LDS
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    p0 += __shfl_down(p0, off);
    p1 += __shfl_down(p1, off);
    p2 += __shfl_down(p2, off);
    p3 += __shfl_down(p3, off);
  }
  __syncthreads();  // partial[] reuse after the dynamics matvec
  if ((tid & 63) == 0) {
    const int wid = tid >> 6;
    partial[wid * 4 + 0] = p0;
    partial[wid * 4 + 1] = p1;
    partial[wid * 4 + 2] = p2;
    partial[wid * 4 + 3] = p3;
  }
  __syncthreads();

  // ---- scalar bookkeeping (thread 0) -------------------------------------
  if (tid == 0) {
    const int nw = nth >> 6;
    float vfwd = 0, vy = 0, h = 0, asq = 0;
    for (int wdx = 0; wdx < nw; ++wdx) {
      vfwd += partial[wdx * 4 + 0];
      vy += partial[wdx * 4 + 1];
      h += partial[wdx * 4 + 2];
      asq += partial[wdx * 4 + 3];
    }
    const float alive_old = w_alive;
    float rew;
    if (la.goal) {
      const float rx = P.goal[(int64_t)b * 2 + 0] - P.pos[(int64_t)b * 3 + 0];
      const float ry = P.goal[(int64_t)b * 2 + 1] - P.pos[(int64_t)b * 3 + 1];
      const float inv = 1.0f / (sqrtf(rx * rx + ry * ry) + 1e-6f);
      rew = vfwd * rx * inv + vy * ry * inv - la.ctrl * asq + la.alive_bonus;
    } else {
      rew = vfwd - la.ctrl * asq + la.alive_bonus;
    }
    const float px = P.pos[(int64_t)b * 3 + 0] + la.dt * vfwd;
    const float py = P.pos[(int64_t)b * 3 + 1] + la.dt * vy;
    P.pos[(int64_t)b * 3 + 0] = px;
    P.pos[(int64_t)b * 3 + 1] = py;
    P.pos[(int64_t)b * 3 + 2] = h;
    const float done = (la.terminate && h < la.fall_thr) ? 1.0f : 0.0f;
    P.rew_total[b] += rew * alive_old;
    P.member_steps[b] += alive_old;
    if (alive_old > 0.0f) {
      P.behv[(int64_t)b * 3 + 0] = px;
      P.behv[(int64_t)b * 3 + 1] = py;
      P.behv[(int64_t)b * 3 + 2] = h;
    }
    P.alive[b] = alive_old * (1.0f - done);
  }
  __syncthreads();

  // goal-relative obs dims of the per-member obs statistics (needs new pos)
  if (la.goal && w_alive > 0.0f && tid < 2) {
    const float rel = (P.goal[(int64_t)b * 2 + tid] - P.pos[(int64_t)b * 3 + tid]) * 0.1f;
    ms[S + tid] += rel;
    mq[S + tid] += rel * rel;
  }
  __syncthreads();  // LDS reuse safety for the episode kernel's next step
}

__device__ __forceinline__ void loco_step_body(
    const MlpShape& sh, const LocoArgs& la, const LocoPtrs& P, int b, uint64_t salt,
    float* bufA, float* bufB, float* partial, float* raws, float* abuf, float* sc) {
  loco_fwd_body(sh, la, P, b, salt, bufA, bufB, partial, raws, abuf);
  if (la.S % 8 == 0)
    loco_dyn_partials(P.Am, la.S, raws, partial, threadIdx.x, blockDim.x);
  loco_dyn_finish(la, P, b, raws, abuf, partial, threadIdx.x, blockDim.x);
}

#define ES_LOCO_CARVE()                                          \
  extern __shared__ __attribute__((aligned(16))) char smem[];    \
  float* bufA = reinterpret_cast<float*>(smem);                  \
  float* bufB = bufA + sh.maxdim;                                \
  float* partial = bufB + sh.maxdim;                             \
  float* raws = partial + 256 * 8;                               \
  float* abuf = raws + ((la.S + 3) & ~3);                        \
  float* sc = abuf + 64;

__global__ void __launch_bounds__(256)
loco_step_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, uint64_t salt) {
  ES_LOCO_CARVE();
  loco_step_body(sh, la, P, blockIdx.x, salt, bufA, bufB, partial, raws, abuf, sc);
}

__global__ void __launch_bounds__(256)
loco_episode_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, int member_base, int n_steps,
                    int salt_base) {
  ES_LOCO_CARVE();
  const int b = member_base + blockIdx.x;
  for (int t = 1; t <= n_steps; ++t)
    loco_step_body(sh, la, P, b, (uint64_t)(salt_base + t), bufA, bufB, partial, raws,
                   abuf, sc);
}

// 512-thread variant for the LATENCY-BOUND noiseless side episode: one
// block runs max_steps sequential steps, so halving every per-step
// partition (PART doubles) shortens the serial dependency chains. Used
// only for the noiseless member (block_threads=512) — the episode-mode
// population grid keeps 256-thread blocks for occupancy.
__global__ void __launch_bounds__(512)
loco_episode512_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, int member_base,
                       int n_steps, int salt_base) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* bufA = reinterpret_cast<float*>(smem);
  float* bufB = bufA + sh.maxdim;
  float* partial = bufB + sh.maxdim;
  float* raws = partial + 512 * 8;
  float* abuf = raws + ((la.S + 3) & ~3);
  float* sc = abuf + 64;
  const int b = member_base + blockIdx.x;
  for (int t = 1; t <= n_steps; ++t)
    loco_step_body(sh, la, P, b, (uint64_t)(salt_base + t), bufA, bufB, partial, raws,
                   abuf, sc);
}

// ---- split-dynamics path ---------------------------------------------------
// The fused kernel re-reads the shared (S, S) bf16 transition matrix A once
// per MEMBER (pop x 283 KB per step for Humanoid = the measured ~20 us/step
// L2-bandwidth bound, profiles/README.md). The split path factors one env
// step into two launches:
//   loco_fwd_kernel  — grid = pop: obs build + policy forward + action
//                      decode, actions written to a (pop, 64) global scratch;
//   loco_dyn_kernel<G> — grid = pop/G: each block stages G members' states
//                      and actions in LDS and FMAs every A octet it loads
//                      into all G members' accumulators, cutting A traffic
//                      G-fold. Per-member accumulation order is IDENTICAL to
//                      the fused kernel (same (oi, ip) tiling, same depth-4
//                      pipeline, same loco_dyn_finish), so split and fused
//                      trajectories match bitwise.
__global__ void __launch_bounds__(256)
loco_fwd_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, uint64_t salt, float* act_glob) {
  ES_LOCO_CARVE();
  (void)sc;
  const int b = blockIdx.x;
  loco_fwd_body(sh, la, P, b, salt, bufA, bufB, partial, raws, abuf);
  if (threadIdx.x < la.A) act_glob[(int64_t)b * 64 + threadIdx.x] = abuf[threadIdx.x];
}

template <int G>
__global__ void __launch_bounds__(256)
loco_dyn_kernel(LocoArgs la, LocoPtrs P, const float* act_glob, int n_pop) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int S = la.S;
  const int Spad = (S + 3) & ~3;
  float* raws = reinterpret_cast<float*>(smem);  // G x Spad raw states
  float* abuf = raws + G * Spad;                 // G x 64 actions
  float* partial = abuf + G * 64;                // G x 2048 matvec partials
  const int tid = threadIdx.x, nth = blockDim.x;
  const int b0 = blockIdx.x * G;
  const int gs = min(G, n_pop - b0);  // members in this block (tail block < G)

  // stage states + actions; zero-fill tail slots so the accumulator loop can
  // stay fully unrolled with compile-time member indices (register arrays)
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const float* sb = P.s_glob + (int64_t)(b0 + g) * S;
    for (int i = tid; i < S; i += nth) raws[g * Spad + i] = g < gs ? sb[i] : 0.0f;
    if (tid < la.A)
      abuf[g * 64 + tid] = g < gs ? act_glob[(int64_t)(b0 + g) * 64 + tid] : 0.0f;
  }
  __syncthreads();

  // shared-A sweep: one uint4 A-octet load feeds G members' bf8_fma chains
  const int OCT = S >> 3;
  const int PART = nth / OCT;
  const int oi = tid % OCT, ip = tid / OCT;
  if (ip < PART) {
    float acc[G][8];
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int q = 0; q < 8; ++q) acc[g][q] = 0.0f;
    const uint16_t* acol = P.Am + (oi << 3);
    auto ld = [&](int i) {
      return *reinterpret_cast<const uint4*>(acol + (int64_t)i * S);
    };
    auto fmaG = [&](const uint4& c, int i) {
#pragma unroll
      for (int g = 0; g < G; ++g) bf8_fma(c, raws[g * Spad + i], acc[g]);
    };
    int i = ip;
    const int step4 = PART * 4;
    if (i + 3 * PART < S) {
      uint4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
      for (; i + 7 * PART < S; i += step4) {
        const uint4 n0 = ld(i + 4 * PART), n1 = ld(i + 5 * PART),
                    n2 = ld(i + 6 * PART), n3 = ld(i + 7 * PART);
        fmaG(c0, i);
        fmaG(c1, i + PART);
        fmaG(c2, i + 2 * PART);
        fmaG(c3, i + 3 * PART);
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      }
      fmaG(c0, i);
      fmaG(c1, i + PART);
      fmaG(c2, i + 2 * PART);
      fmaG(c3, i + 3 * PART);
      i += step4;
    }
    for (; i < S; i += PART) fmaG(ld(i), i);
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int q = 0; q < 8; ++q)
        partial[g * 2048 + (ip * OCT + oi) * 8 + q] = acc[g][q];
  }
  __syncthreads();

  for (int g = 0; g < gs; ++g)
    loco_dyn_finish(la, P, b0 + g, raws + g * Spad, abuf + g * 64,
                    partial + g * 2048, tid, nth);
}

template <int G>
static int loco_launch_dyn(const LocoArgs& la, const LocoPtrs& P, const float* act_glob,
                           int n_pop, hipStream_t stream) {
  const int Spad = (la.S + 3) & ~3;
  const unsigned lds = (unsigned)((G * Spad + G * 64 + G * 2048) * sizeof(float));
  const unsigned grid = (unsigned)((n_pop + G - 1) / G);
  loco_dyn_kernel<G><<<dim3(grid), dim3(256), lds, stream>>>(la, P, act_glob, n_pop);
  ES_CHECK_LAUNCH();
  return 0;
}

// ---- antithetic-pair rollout step ------------------------------------------
// One block per (pair, episode): the +noise and -noise members' forwards run
// together off ONE HBM sigma*eps stream plus the L2-resident shared theta
// stream (mlp_layers_pair, mlp_core.h) — the per-step HBM weight traffic
// halves vs materialized per-member blobs. Dynamics shares each A octet
// between the two members; the per-member epilogue is loco_dyn_finish
// verbatim, so per-slot bookkeeping is identical to the fused step.
// Effective weights are bf16(theta) +- bf16(sigma*eps) (two roundings); with
// sigma = 0 the trajectories are BITWISE-identical to es_loco_step.

__device__ __forceinline__ void loco_dyn_partials_pair(
    const uint16_t* Am, int S, const float* rawsP, const float* rawsM,
    float* partP, float* partM, int tid, int nth) {
  const int OCT = S >> 3;
  const int PART = nth / OCT;
  const int oi = tid % OCT, ip = tid / OCT;
  float accp[8], accm[8];
#pragma unroll
  for (int q = 0; q < 8; ++q) accp[q] = accm[q] = 0.0f;
  if (ip < PART) {
    const uint16_t* acol = Am + (oi << 3);
    auto ld = [&](int i) {
      return *reinterpret_cast<const uint4*>(acol + (int64_t)i * S);
    };
    auto fma2 = [&](const uint4& c, int i) {
      bf8_fma(c, rawsP[i], accp);
      bf8_fma(c, rawsM[i], accm);
    };
    int i = ip;
    const int step4 = PART * 4;
#if ES_DYN_DEPTH8
    if (i + 7 * PART < S) {
      uint4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
      uint4 d0 = ld(i + 4 * PART), d1 = ld(i + 5 * PART), d2 = ld(i + 6 * PART),
            d3 = ld(i + 7 * PART);
      for (; i + 11 * PART < S; i += step4) {
        const uint4 n0 = ld(i + 8 * PART), n1 = ld(i + 9 * PART),
                    n2 = ld(i + 10 * PART), n3 = ld(i + 11 * PART);
        fma2(c0, i);
        fma2(c1, i + PART);
        fma2(c2, i + 2 * PART);
        fma2(c3, i + 3 * PART);
        c0 = d0; c1 = d1; c2 = d2; c3 = d3;
        d0 = n0; d1 = n1; d2 = n2; d3 = n3;
      }
      // drain the 8 preloaded blocks (guarded; order stays ascending)
      if (i < S) fma2(c0, i);
      if (i + PART < S) fma2(c1, i + PART);
      if (i + 2 * PART < S) fma2(c2, i + 2 * PART);
      if (i + 3 * PART < S) fma2(c3, i + 3 * PART);
      if (i + 4 * PART < S) fma2(d0, i + 4 * PART);
      if (i + 5 * PART < S) fma2(d1, i + 5 * PART);
      if (i + 6 * PART < S) fma2(d2, i + 6 * PART);
      if (i + 7 * PART < S) fma2(d3, i + 7 * PART);
      i += 8 * PART;
    }
#else
    if (i + 3 * PART < S) {
      uint4 c0 = ld(i), c1 = ld(i + PART), c2 = ld(i + 2 * PART), c3 = ld(i + 3 * PART);
      for (; i + 7 * PART < S; i += step4) {
        const uint4 n0 = ld(i + 4 * PART), n1 = ld(i + 5 * PART),
                    n2 = ld(i + 6 * PART), n3 = ld(i + 7 * PART);
        fma2(c0, i);
        fma2(c1, i + PART);
        fma2(c2, i + 2 * PART);
        fma2(c3, i + 3 * PART);
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      }
      fma2(c0, i);
      fma2(c1, i + PART);
      fma2(c2, i + 2 * PART);
      fma2(c3, i + 3 * PART);
      i += step4;
    }
#endif
    for (; i < S; i += PART) fma2(ld(i), i);
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      partP[(ip * OCT + oi) * 8 + q] = accp[q];
      partM[(ip * OCT + oi) * 8 + q] = accm[q];
    }
  }
  __syncthreads();
}

// ES_DYN_EARLY (experiment knob): compute the A-matvec partials BEFORE the
// policy forward, into a dedicated LDS slab — the A sweep needs only the
// raw state, so its L2 latency then overlaps the forward's HBM weight
// streaming deterministically within the block (instead of statistically
// across blocks). Costs 16 KB LDS per block (one resident block at fp8's
// 4-block occupancy); which wins is measured.
#ifndef ES_DYN_EARLY
#define ES_DYN_EARLY 0
#endif

// EB = uint16_t (bf16 eps rows) or uint8_t (fp8 row-pair-interleaved rows)
template <typename EB>
__device__ __forceinline__ void loco_pair_step_body(
    const MlpShape& sh, const LocoArgs& la, const LocoPtrs& P,
    const uint16_t* tb, const EB* eb, int bp, int bm, uint64_t salt,
    float* bufAp, float* bufAm, float* bufBp, float* bufBm, float* partial,
    float* rawsP, float* rawsM, float* abufP, float* abufM,
    float* partA = nullptr) {
  const int tid = threadIdx.x, nth = blockDim.x;
  loco_build_obs(la, P, bp, bufAp, rawsP, tid, nth);
  loco_build_obs(la, P, bm, bufAm, rawsM, tid, nth);
  __syncthreads();
  const bool early = ES_DYN_EARLY && partA != nullptr && la.S % 8 == 0;
  if (early)
    loco_dyn_partials_pair(P.Am, la.S, rawsP, rawsM, partA, partA + 2048,
                           tid, nth);
  float *ap, *am;
  if constexpr (sizeof(EB) == 1)
    mlp_layers_pair_fp8(tb, (const uint8_t*)eb, sh, bufAp, bufAm, bufBp, bufBm,
                        partial, tid, nth, 1, &ap, &am);
  else
    mlp_layers_pair(tb, (const uint16_t*)eb, sh, bufAp, bufAm, bufBp, bufBm,
                    partial, tid, nth, 1, &ap, &am);
  loco_decode_action(sh, la, P, bp, salt, ap, abufP, tid);
  loco_decode_action(sh, la, P, bm, salt, am, abufM, tid);
  __syncthreads();
  float* pd = early ? partA : partial;
  if (!early && la.S % 8 == 0)
    loco_dyn_partials_pair(P.Am, la.S, rawsP, rawsM, partial, partial + 2048,
                           tid, nth);
  loco_dyn_finish(la, P, bp, rawsP, abufP, pd, tid, nth);
  loco_dyn_finish(la, P, bm, rawsM, abufM, pd + 2048, tid, nth);
}

#define ES_LOCO_PAIR_CARVE()                                     \
  extern __shared__ __attribute__((aligned(16))) char smem[];    \
  float* bufAp = reinterpret_cast<float*>(smem);                 \
  float* bufAm = bufAp + sh.maxdim;                              \
  float* bufBp = bufAm + sh.maxdim;                              \
  float* bufBm = bufBp + sh.maxdim;                              \
  float* partial = bufBm + sh.maxdim; /* 2 x 256*8 */            \
  float* rawsP = partial + 2 * 256 * 8;                          \
  float* rawsM = rawsP + ((la.S + 3) & ~3);                      \
  float* abufP = rawsM + ((la.S + 3) & ~3);                      \
  float* abufM = abufP + 64;                                     \
  float* partA = ES_DYN_EARLY ? abufM + 64 : nullptr;            \
  const int q = blockIdx.x;                                      \
  const int p = q / la.eps, e = q % la.eps;                      \
  const int bp = p * la.eps + e;                                 \
  const int bm = (n_pairs + p) * la.eps + e;                     \
  const auto* ebp = eb + (int64_t)p * la.row_stride;

// ES_PAIR_MINWAVES (experiment knob): force a min-waves/SIMD bound on the
// pair step kernel so deeper rings can be capped to 3 waves (168 VGPRs)
#ifndef ES_PAIR_MINWAVES
#define ES_PAIR_MINWAVES 1
#endif
__global__ void __launch_bounds__(256, ES_PAIR_MINWAVES)
loco_pair_step_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, const uint16_t* tb,
                      const uint16_t* eb, int n_pairs, uint64_t salt) {
  ES_LOCO_PAIR_CARVE();
  loco_pair_step_body(sh, la, P, tb, ebp, bp, bm, salt, bufAp, bufAm, bufBp, bufBm,
                      partial, rawsP, rawsM, abufP, abufM, partA);
}

__global__ void __launch_bounds__(256, ES_PAIR_MINWAVES)
loco_pair_step_fp8_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, const uint16_t* tb,
                          const uint8_t* eb, int n_pairs, uint64_t salt) {
  ES_LOCO_PAIR_CARVE();
  loco_pair_step_body(sh, la, P, tb, ebp, bp, bm, salt, bufAp, bufAm, bufBp, bufBm,
                      partial, rawsP, rawsM, abufP, abufM, partA);
}

// Whole-episode (or k-step chunk) pair rollout in ONE launch: each block
// owns its (pair, episode) slot for n_steps consecutive env steps. Unlike
// the per-step launch, blocks never rendezvous chip-wide, so one block's
// L2-side dynamics phase overlaps other blocks' HBM sigma*eps streaming
// (the per-step grid runs phase-locked: every block starts its forward at
// launch, leaving HBM idle during the correlated dynamics phases — measured
// 47% HBM duty at the flagship config). Trajectories are bitwise-identical
// to n_steps sequential es_loco_pair_step launches (same salt_base + t
// sequence, same body).
// unroll(disable): the step loop's only carried state is scalar, but an
// unrolled body doubles the live vector ranges (256 VGPRs = 2 blocks/CU,
// or 328 B/lane spills under a 3-wave bound). Episode mode REQUIRES
// >= 3 blocks/CU: 640 flagship blocks must all be resident, since blocks
// retire only at episode end.
__global__ void __launch_bounds__(256, 4)
loco_pair_episode_kernel(MlpShape sh, LocoArgs la, LocoPtrs P, const uint16_t* tb,
                         const uint16_t* eb, int n_pairs, int n_steps,
                         int salt_base) {
  ES_LOCO_PAIR_CARVE();
#pragma clang loop unroll(disable)
  for (int t = 1; t <= n_steps; ++t)
    loco_pair_step_body(sh, la, P, tb, ebp, bp, bm, (uint64_t)(salt_base + t),
                        bufAp, bufAm, bufBp, bufBm, partial, rawsP, rawsM, abufP,
                        abufM, partA);
}

// (A 128-thread 2-waves/SIMD variant — zero spills, 4 blocks/CU — was
// measured 24% SLOWER than per-step launches and cannot match them bitwise
// (different PART split changes summation order); removed after the A/B.)

static int loco_prepare(MlpShape* sh, LocoArgs* la, const int32_t* dims_host, int32_t ndims,
                        int64_t row_stride, float ob_clip, int32_t sdim, int32_t adim,
                        int32_t goal_flag, int32_t terminate, int32_t noiseless_from,
                        int32_t bins, int32_t eps, int32_t act_mode, float leak,
                        float ctrl, float alive_bonus, float fall_thr, float dt,
                        unsigned* lds) {
  int rc = mlp_shape_init(sh, dims_host, ndims, row_stride);
  if (rc) return rc;
  if (adim > 64 || sdim > ES_MAXDIM) return -103;
  if (sh->dims[0] != sdim + (goal_flag ? 2 : 0)) return -104;
  int out_dim = adim;
  if (bins > 1) out_dim = adim * bins;
  else if (act_mode == 2) out_dim = adim + 1;
  else if (act_mode == 3) out_dim = adim * 2;
  if (sh->dims[sh->n_layers] != out_dim) return -105;
  la->S = sdim; la->A = adim; la->D = sh->dims[0]; la->goal = goal_flag;
  la->terminate = terminate; la->noiseless_from = noiseless_from; la->bins = bins;
  la->act_mode = act_mode;
  la->eps = eps > 0 ? eps : 1;
  la->wrow0 = 0;
  la->leak = leak; la->ctrl = ctrl; la->alive_bonus = alive_bonus; la->fall_thr = fall_thr;
  la->dt = dt; la->ob_clip = ob_clip; la->row_stride = row_stride;
  *lds = (unsigned)(mlp_lds_bytes(sh->maxdim) + (((sdim + 3) & ~3) + 64 + 8) * 4);
  return 0;
}

static LocoPtrs loco_ptrs(const void* weights, const void* obmean, const void* obstd,
                          const void* ac_std_dev, const void* seed_dev, void* s_glob,
                          void* pos, const void* goal, const void* Am, const void* Bm,
                          const void* b0, const void* wv, const void* wa, const void* wy,
                          const void* wh, void* alive, void* rew_total, void* member_steps,
                          void* behv, void* mo_sum, void* mo_sumsq) {
  LocoPtrs P;
  P.weights = (const uint16_t*)weights;
  P.obmean = (const float*)obmean; P.obstd = (const float*)obstd;
  P.ac_std_dev = (const float*)ac_std_dev; P.seed_dev = (const uint64_t*)seed_dev;
  P.s_glob = (float*)s_glob; P.pos = (float*)pos; P.goal = (const float*)goal;
  P.Am = (const uint16_t*)Am; P.Bm = (const float*)Bm; P.b0 = (const float*)b0;
  P.wv = (const float*)wv; P.wa = (const float*)wa; P.wy = (const float*)wy;
  P.wh = (const float*)wh; P.alive = (float*)alive; P.rew_total = (float*)rew_total;
  P.member_steps = (float*)member_steps; P.behv = (float*)behv;
  P.mo_sum = (float*)mo_sum; P.mo_sumsq = (float*)mo_sumsq;
  return P;
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
                            int32_t eps, int32_t act_mode, float leak, float ctrl,
                            float alive_bonus, float fall_thr, float dt, void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds);
  if (rc) return rc;
  LocoPtrs P = loco_ptrs(weights, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos, goal,
                         Am, Bm, b0, wv, wa, wy, wh, alive, rew_total, member_steps, behv,
                         mo_sum, mo_sumsq);
  loco_step_kernel<<<dim3((unsigned)n_pop), dim3(256), lds, (hipStream_t)stream>>>(
      sh, la, P, salt);
  ES_CHECK_LAUNCH();
  return 0;
}

// Split-dynamics step: forward kernel (grid = pop) then shared-A dynamics
// kernel (grid = pop/G). act_glob = (n_pop, 64) float32 scratch. Requires
// S % 8 == 0 (octet A tiling) and G in {2, 4, 5, 8}. Bitwise-identical
// trajectories to es_loco_step (tests/test_gpu_kernels.py).
extern "C" int es_loco_step_split(
    const void* weights, const void* obmean, const void* obstd,
    const int32_t* dims_host, int32_t ndims, const void* seed_dev, uint64_t salt,
    float ob_clip, const void* ac_std_dev, int64_t row_stride, void* s_glob, void* pos,
    const void* goal, const void* Am, const void* Bm, const void* b0, const void* wv,
    const void* wa, const void* wy, const void* wh, void* alive, void* rew_total,
    void* member_steps, void* behv, void* mo_sum, void* mo_sumsq, int32_t n_pop,
    int32_t sdim, int32_t adim, int32_t goal_flag, int32_t terminate,
    int32_t noiseless_from, int32_t bins, int32_t eps, int32_t act_mode, float leak,
    float ctrl, float alive_bonus, float fall_thr, float dt, void* act_glob, int32_t G,
    void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds);
  if (rc) return rc;
  if (sdim % 8 != 0) return -106;
  LocoPtrs P = loco_ptrs(weights, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos, goal,
                         Am, Bm, b0, wv, wa, wy, wh, alive, rew_total, member_steps, behv,
                         mo_sum, mo_sumsq);
  hipStream_t s = (hipStream_t)stream;
  loco_fwd_kernel<<<dim3((unsigned)n_pop), dim3(256), lds, s>>>(sh, la, P, salt,
                                                                (float*)act_glob);
  ES_CHECK_LAUNCH();
  switch (G) {
    case 2: return loco_launch_dyn<2>(la, P, (const float*)act_glob, n_pop, s);
    case 4: return loco_launch_dyn<4>(la, P, (const float*)act_glob, n_pop, s);
    case 5: return loco_launch_dyn<5>(la, P, (const float*)act_glob, n_pop, s);
    case 8: return loco_launch_dyn<8>(la, P, (const float*)act_glob, n_pop, s);
    default: return -107;
  }
}

// Antithetic-pair step: theta_row = (1, row_stride) bf16(theta);
// eps_rows = (n_pairs, row_stride) bf16(sigma*eps). Grid = n_pairs * eps
// blocks, each evaluating the +/- slots of one (pair, episode).
extern "C" int es_loco_pair_step(
    const void* theta_row, const void* eps_rows, const void* obmean,
    const void* obstd, const int32_t* dims_host, int32_t ndims, const void* seed_dev,
    uint64_t salt, float ob_clip, const void* ac_std_dev, int64_t row_stride,
    void* s_glob, void* pos, const void* goal, const void* Am, const void* Bm,
    const void* b0, const void* wv, const void* wa, const void* wy, const void* wh,
    void* alive, void* rew_total, void* member_steps, void* behv, void* mo_sum,
    void* mo_sumsq, int32_t n_pairs, int32_t sdim, int32_t adim, int32_t goal_flag,
    int32_t terminate, int32_t noiseless_from, int32_t bins, int32_t eps,
    int32_t act_mode, float leak, float ctrl, float alive_bonus, float fall_thr,
    float dt, void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds_unused;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds_unused);
  if (rc) return rc;
  LocoPtrs P = loco_ptrs(nullptr, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos,
                         goal, Am, Bm, b0, wv, wa, wy, wh, alive, rew_total,
                         member_steps, behv, mo_sum, mo_sumsq);
  const int Spad = (sdim + 3) & ~3;
  const unsigned lds =
      (unsigned)((4 * sh.maxdim + 2 * 256 * 8 + 2 * Spad + 2 * 64 +
                  (ES_DYN_EARLY ? 2 * 2048 : 0)) * sizeof(float));
  const unsigned grid = (unsigned)(n_pairs * la.eps);
  loco_pair_step_kernel<<<dim3(grid), dim3(256), lds, (hipStream_t)stream>>>(
      sh, la, P, (const uint16_t*)theta_row, (const uint16_t*)eps_rows, n_pairs, salt);
  ES_CHECK_LAUNCH();
  return 0;
}

// fp8-eps pair step: eps_rows = (n_pairs, row_stride) BYTES, written by
// es_pheno_fp8 (row-pair-interleaved e4m3); theta_row stays bf16.
extern "C" int es_loco_pair_step_fp8(
    const void* theta_row, const void* eps_rows, const void* obmean,
    const void* obstd, const int32_t* dims_host, int32_t ndims, const void* seed_dev,
    uint64_t salt, float ob_clip, const void* ac_std_dev, int64_t row_stride,
    void* s_glob, void* pos, const void* goal, const void* Am, const void* Bm,
    const void* b0, const void* wv, const void* wa, const void* wy, const void* wh,
    void* alive, void* rew_total, void* member_steps, void* behv, void* mo_sum,
    void* mo_sumsq, int32_t n_pairs, int32_t sdim, int32_t adim, int32_t goal_flag,
    int32_t terminate, int32_t noiseless_from, int32_t bins, int32_t eps,
    int32_t act_mode, float leak, float ctrl, float alive_bonus, float fall_thr,
    float dt, void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds_unused;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds_unused);
  if (rc) return rc;
  LocoPtrs P = loco_ptrs(nullptr, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos,
                         goal, Am, Bm, b0, wv, wa, wy, wh, alive, rew_total,
                         member_steps, behv, mo_sum, mo_sumsq);
  const int Spad = (sdim + 3) & ~3;
  const unsigned lds =
      (unsigned)((4 * sh.maxdim + 2 * 256 * 8 + 2 * Spad + 2 * 64 +
                  (ES_DYN_EARLY ? 2 * 2048 : 0)) * sizeof(float));
  const unsigned grid = (unsigned)(n_pairs * la.eps);
  loco_pair_step_fp8_kernel<<<dim3(grid), dim3(256), lds, (hipStream_t)stream>>>(
      sh, la, P, (const uint16_t*)theta_row, (const uint8_t*)eps_rows, n_pairs, salt);
  ES_CHECK_LAUNCH();
  return 0;
}

// Pair-episode: n_steps consecutive env steps per launch (n_steps =
// max_steps -> whole generation in one launch; smaller -> chunked step
// mode). Same buffers as es_loco_pair_step.
extern "C" int es_loco_pair_episode(
    const void* theta_row, const void* eps_rows, const void* obmean,
    const void* obstd, const int32_t* dims_host, int32_t ndims, const void* seed_dev,
    float ob_clip, const void* ac_std_dev, int64_t row_stride,
    void* s_glob, void* pos, const void* goal, const void* Am, const void* Bm,
    const void* b0, const void* wv, const void* wa, const void* wy, const void* wh,
    void* alive, void* rew_total, void* member_steps, void* behv, void* mo_sum,
    void* mo_sumsq, int32_t n_pairs, int32_t sdim, int32_t adim, int32_t goal_flag,
    int32_t terminate, int32_t noiseless_from, int32_t bins, int32_t eps,
    int32_t act_mode, float leak, float ctrl, float alive_bonus, float fall_thr,
    float dt, int32_t n_steps, int32_t salt_base, void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds_unused;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds_unused);
  if (rc) return rc;
  LocoPtrs P = loco_ptrs(nullptr, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos,
                         goal, Am, Bm, b0, wv, wa, wy, wh, alive, rew_total,
                         member_steps, behv, mo_sum, mo_sumsq);
  const int Spad = (sdim + 3) & ~3;
  const unsigned lds =
      (unsigned)((4 * sh.maxdim + 2 * 256 * 8 + 2 * Spad + 2 * 64 +
                  (ES_DYN_EARLY ? 2 * 2048 : 0)) * sizeof(float));
  const unsigned grid = (unsigned)(n_pairs * la.eps);
  loco_pair_episode_kernel<<<dim3(grid), dim3(256), lds, (hipStream_t)stream>>>(
      sh, la, P, (const uint16_t*)theta_row, (const uint16_t*)eps_rows, n_pairs,
      n_steps, salt_base);
  ES_CHECK_LAUNCH();
  return 0;
}

extern "C" int es_loco_episode(const void* weights, const void* obmean, const void* obstd,
                               const int32_t* dims_host, int32_t ndims, const void* seed_dev,
                               int32_t n_steps, float ob_clip, const void* ac_std_dev,
                               int64_t row_stride,
                               void* s_glob, void* pos, const void* goal, const void* Am,
                               const void* Bm, const void* b0, const void* wv,
                               const void* wa, const void* wy, const void* wh, void* alive,
                               void* rew_total, void* member_steps, void* behv,
                               void* mo_sum, void* mo_sumsq, int32_t member_base,
                               int32_t n_members, int32_t salt_base, int32_t wrow0,
                               int32_t sdim, int32_t adim,
                               int32_t goal_flag, int32_t terminate, int32_t noiseless_from,
                               int32_t bins, int32_t eps, int32_t act_mode, float leak,
                               float ctrl, float alive_bonus, float fall_thr, float dt,
                               int32_t block_threads, void* stream) {
  MlpShape sh;
  LocoArgs la;
  unsigned lds;
  int rc = loco_prepare(&sh, &la, dims_host, ndims, row_stride, ob_clip, sdim, adim,
                        goal_flag, terminate, noiseless_from, bins, eps, act_mode, leak,
                        ctrl, alive_bonus, fall_thr, dt, &lds);
  if (rc) return rc;
  la.wrow0 = wrow0;
  LocoPtrs P = loco_ptrs(weights, obmean, obstd, ac_std_dev, seed_dev, s_glob, pos, goal,
                         Am, Bm, b0, wv, wa, wy, wh, alive, rew_total, member_steps, behv,
                         mo_sum, mo_sumsq);
  if (block_threads == 512) {
    const unsigned lds512 = lds + 256 * 8 * 4;  // partial[] grows with nth
    loco_episode512_kernel<<<dim3((unsigned)n_members), dim3(512), lds512,
                             (hipStream_t)stream>>>(sh, la, P, member_base, n_steps,
                                                    salt_base);
  } else {
    loco_episode_kernel<<<dim3((unsigned)n_members), dim3(256), lds,
                          (hipStream_t)stream>>>(sh, la, P, member_base, n_steps,
                                                 salt_base);
  }
  ES_CHECK_LAUNCH();
  return 0;
}
