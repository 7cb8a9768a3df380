"""GpuEngine — whole-generation population-batched ES on one GPU per rank.

This is the MI355X-native replacement for the reference's hot loop
(``src/core/es.py:54-81``: E sequential pheno+rollout pairs per CPU process).
One rank = one GPU; per generation:

1. sample the rank's noise offsets (numpy RandomState, reference-compatible
   index semantics, ``noisetable.py:37-40``) — one batched draw;
2. HIP pheno kernel materializes the member parameters in HBM — either as
   per-member bf16 blobs [+noise pairs | -noise pairs | noiseless slot], or
   (pair_rollout, the flagship default) as ONE bf16 theta row plus per-pair
   sigma*eps rows — fp8 e4m3 on supporting layouts (bytes AND load count
   halve; the gradient gather re-quantizes through the same converters so
   the update is estimator-exact), bf16 otherwise — that the pair rollout
   kernel combines into W+/- in registers (the noiseless evaluation of
   ``es.py:48`` rides in the same batch either way);
3. the rollout loop — fused HIP MLP forward over the whole population +
   batched env dynamics — is captured once into a hipGraph and replayed
   every generation (launch-overhead-free inner loop);
   finished episodes stay in the batch under an ``alive`` mask with their
   rewards/behaviour frozen, matching the reference's break-on-done
   (``gym_runner.py:63-64``);
4. (fit+, fit-, idx) fp64 triples go through ONE RCCL all_gather over xGMI
   (replaces the replicated Alltoall, ``es.py:84-95``); ranking runs
   redundantly on every rank (reference README.md:10-12 — parameters are
   never communicated);
5. the HIP gather-GEMV reconstructs the gradient straight from the HBM noise
   table and the fused Adam/SGD kernel updates theta in place.

Parameter layout: device theta lives in FORWARD layout (per layer: W^T, b)
so pheno/forward/grad are all coalesced; a precomputed permutation maps it
to the reference's flat state_dict layout at the host boundary (checkpoint
save/load keeps the reference's pickle format, ``policy.py:37-47``).
"""
from __future__ import annotations

import time
from typing import List, Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd import ops
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs.base import BatchedEnv
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import SGD, Adam, Optimizer
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import RewardResult
from es_pytorch_amd.utils.novelty import novelty_batch
from es_pytorch_amd.utils.rankers import CenteredRanker, Ranker


def forward_perm(dims: List[int]) -> torch.Tensor:
    """Map forward-layout element index -> flat state_dict index.

    flat (reference ``policy.py:33-35``): per layer, W (O, I) row-major then
    b (O). forward: per layer, W^T (I, O) row-major then b (O).
    """
    perm = []
    flat_off = 0
    for I, O in zip(dims[:-1], dims[1:]):
        w = np.arange(O * I, dtype=np.int64).reshape(O, I)  # flat W indices
        perm.append((flat_off + w.T).reshape(-1))           # (I, O) order
        flat_off += O * I
        perm.append(np.arange(O, dtype=np.int64) + flat_off)
        flat_off += O
    return torch.from_numpy(np.concatenate(perm))


class _NoiselessResult(RewardResult):
    """Adapter presenting the engine's noiseless slot to the Reporter API."""

    def __init__(self, reward: float, position, steps: int, ob_shape):
        super().__init__([reward], [float(position[0]), float(position[1]),
                                    float(position[2])] * 2,
                         np.zeros((1,) + tuple(ob_shape)), steps)


class GpuEngine:
    def __init__(self, cfg, comm: Comm, policy: Policy, nt: NoiseTable, env: BatchedEnv,
                 rs: np.random.RandomState, objective: str = "reward",
                 use_graph: bool = True, novelty_k: int = 10,
                 fused: Optional[bool] = None, rollout_mode: str = "step",
                 split_dyn: Optional[bool] = None,
                 pair_rollout: Optional[bool] = None,
                 eps_fp8: Optional[bool] = None):
        # rollout_mode: "step" = one kernel per env step for the population
        # (graph-replayed); "episode" = ONE kernel per generation, each block
        # runs its member's whole episode (members are mutually independent,
        # so no per-step rendezvous is required)
        self.rollout_mode = rollout_mode
        # steps per launch in "step" mode: k>1 runs k consecutive env steps
        # per kernel launch (bounded block drift, fewer launch boundaries)
        self.steps_per_launch = int(cfg.general.get("steps_per_launch", 1) or 1)
        self.cfg = cfg
        self.comm = comm
        self.policy = policy
        self.nt = nt
        self.env = env
        self.rs = rs
        self.objective = objective
        self.device = nt.noise.device
        self.use_graph = use_graph and self.device.type == "cuda"
        self.novelty_k = novelty_k
        self.archive: Optional[torch.Tensor] = None  # (N, 2) device tensor for ns/nsr

        ppg = cfg.general.policies_per_gen
        assert ppg % comm.size == 0 and (ppg / comm.size) % 2 == 0, \
            "policies_per_gen must split into antithetic pairs per rank"
        self.pairs = int(ppg // comm.size // 2)
        # episodes per perturbation (reference obj.py:56-63 averaging)
        self.eps = max(1, int(cfg.general.get("eps_per_policy", 1) or 1))
        self.M = 2 * self.pairs + 1      # members: [+pairs | -pairs | noiseless]
        self.B = self.M * self.eps       # evaluation slots (env batch)
        assert env.batch == self.B, f"env batch {env.batch} != engine batch {self.B}"

        self.dims = policy._module.layer_dims()
        # config-time surface of the HIP kernels' compile-time shape limits
        # (mlp_core.h ES_MAXL/ES_MAXDIM) — fail before any launch, not with a
        # kernel error code mid-generation
        if len(self.dims) - 1 > 8:
            raise ValueError(
                f"model has {len(self.dims) - 1} layers; the fused HIP forward "
                f"supports at most 8 (ES_MAXL in ops/csrc/hip/mlp_core.h) — "
                f"reduce policy.layer_sizes or raise ES_MAXL and rebuild")
        if max(self.dims) > 2048:
            raise ValueError(
                f"widest layer is {max(self.dims)}; the fused HIP forward "
                f"supports at most 2048 (ES_MAXDIM in ops/csrc/hip/mlp_core.h) "
                f"— narrow policy.layer_sizes or raise ES_MAXDIM and rebuild")
        # K9: FFBinned policies emit adim*bins logits decoded in-kernel
        self.bins = int(getattr(policy._module, "bins", 0))
        # integrated-gaussian-action variants (reference nn.py:53-96): the net
        # emits its own action std — mode 2 (first output) / 3 (second half).
        # NOTE: the engine contract sizes the output layer adim+1 / 2*adim.
        from es_pytorch_amd.nn.nn import FFIntegGausAction, FFIntegGausActionMulti
        if isinstance(policy._module, FFIntegGausActionMulti):
            self.act_mode = 3
        elif isinstance(policy._module, FFIntegGausAction):
            self.act_mode = 2
        else:
            self.act_mode = 0
        # single output-layer contract across engine and episodic paths
        # (nn.py _ActionView): verify the module was sized for THIS env
        out = self.dims[-1]
        expect = {0: env.ac_dim if self.bins <= 1 else env.ac_dim * self.bins,
                  2: env.ac_dim + 1, 3: 2 * env.ac_dim}[self.act_mode]
        if out != expect:
            raise ValueError(
                f"output layer is {out} wide but act_mode {self.act_mode} on a "
                f"{env.ac_dim}-action env needs {expect} — was the module built "
                f"for a different env?")
        self.n = int(np.sum([I * O + O for I, O in zip(self.dims[:-1], self.dims[1:])]))
        assert self.n == len(policy), (self.n, len(policy))
        self.perm = forward_perm(self.dims).to(self.device)  # fwd idx -> flat idx
        self.dims_arr = (np.array(self.dims, dtype=np.int32))

        d = self.device
        flat = torch.from_numpy(policy.flat_params).to(d)
        self.theta = flat[self.perm].contiguous()            # forward layout fp32
        self.m = torch.zeros(self.n, dtype=torch.float32, device=d)
        self.v = torch.zeros(self.n, dtype=torch.float32, device=d)
        self._load_optim_state()

        # member blob rows padded to 16 B so the forward kernel's uint4
        # (8 x bf16) vector loads stay aligned for every member
        self.row_stride = (self.n + 7) // 8 * 8
        self.weights = torch.empty((self.M, self.row_stride), dtype=torch.bfloat16, device=d)
        self.offsets = torch.zeros(self.M, dtype=torch.int64, device=d)
        self.signs = torch.cat([torch.ones(self.pairs), -torch.ones(self.pairs),
                                torch.zeros(1)]).to(d)
        self.grad = torch.empty(self.n, dtype=torch.float32, device=d)
        self.seed_dev = torch.zeros(1, dtype=torch.int64, device=d)
        # device-resident mutable scalars read by kernels (graph-replay-safe)
        self.acstd_dev = torch.zeros(1, dtype=torch.float32, device=d)

        if self.bins > 1:
            self.alow_dev = torch.from_numpy(
                np.asarray(env.action_space.low, dtype=np.float32)).to(d)
            self.arange_dev = torch.from_numpy(np.asarray(
                env.action_space.high - env.action_space.low, dtype=np.float32)).to(d)
        else:
            self.alow_dev = self.arange_dev = None

        D = env.ob_dim
        self.obmean = torch.zeros(D, dtype=torch.float32, device=d)
        self.obstd = torch.ones(D, dtype=torch.float32, device=d)
        self._push_obstat()

        # rollout state buffers (persistent; written in-place inside the graph)
        self.actions = torch.empty((self.B, env.ac_dim), dtype=torch.float32, device=d)
        self.alive = torch.ones(self.B, dtype=torch.float32, device=d)
        self.rew_total = torch.zeros(self.B, dtype=torch.float32, device=d)
        self.member_steps = torch.zeros(self.B, dtype=torch.float32, device=d)
        self.behv = torch.zeros((self.B, 3), dtype=torch.float32, device=d)
        self.save_mask = torch.zeros(self.B, dtype=torch.float32, device=d)
        self.ob_sum = torch.zeros(D, dtype=torch.float64, device=d)
        self.ob_sumsq = torch.zeros(D, dtype=torch.float64, device=d)
        self.ob_count = torch.zeros((), dtype=torch.float64, device=d)
        self.obs_buf = torch.zeros((self.B, D), dtype=torch.float32, device=d)

        self.max_steps = int(cfg.env.max_steps)
        self.gen = 0
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self.timings = {}
        # side stream: next generation's noise-offset upload overlaps the
        # fitness all-gather (BASELINE north star)
        self._side = torch.cuda.Stream(self.device) if self.device.type == "cuda" else None
        self._offs_prefetched = False

        # fused single-kernel rollout step for the locomotion envs: policy
        # forward + dynamics + bookkeeping in one launch per step
        from es_pytorch_amd.envs.locomotion import SyntheticLocomotion
        self.fused = (isinstance(env, SyntheticLocomotion) and self.device.type == "cuda"
                      if fused is None else bool(fused))
        if self.fused:
            self.mo_sum = torch.zeros((self.B, D), dtype=torch.float32, device=d)
            self.mo_sumsq = torch.zeros((self.B, D), dtype=torch.float32, device=d)

        # split-dynamics rollout: forward kernel + shared-A dynamics kernel
        # per step (G members per dynamics block re-use each A octet load);
        # bitwise-identical trajectories to the fused single-kernel step
        self.dyn_group = int(cfg.general.get("dyn_group", 5) or 5)
        want_split = bool(cfg.general.get("split_dyn", False)) if split_dyn is None \
            else bool(split_dyn)
        self.split_dyn = (want_split and self.fused and env.sdim % 8 == 0
                          and env.ac_dim <= 64 and self.dyn_group in (2, 4, 5, 8))
        if self.split_dyn:
            self.act_scratch = torch.empty(self.B * 64, dtype=torch.float32, device=d)

        # antithetic-pair rollout: the +/- members of a pair share one HBM
        # sigma*eps stream plus the L2-resident shared theta stream
        # (mlp_layers_pair) — per-step HBM weight traffic halves vs the
        # materialized per-member blobs. Effective weights become
        # bf16(theta) +- bf16(sigma*eps) (two bf16 roundings instead of one).
        # constructor True FORCES pair (tests, explicit intent); the config
        # flag is auto-gated on grid size: pair halves the block count, so
        # tiny populations would idle CUs (Hopper pop 256 -> 128 pair blocks
        # measured 9.0M vs 12.4M env-steps/s fused bf16). Crossovers are
        # measured per eps encoding: bf16 pair wins from ~384 pair blocks
        # (pop 512 fused 58.0 vs pair 58.9 us/step, 768 73.3 vs 70.3, 1024
        # 89.6 vs 75.2); fp8 pair from ~256 blocks (round 2, same-box: pop
        # 512 fused 53.5 vs pair-fp8 50.1, 640 61.2 vs 54.7, 768 69.0 vs
        # 56.3, 1024 84.4 vs 59.1 — the halved load count pays earlier).
        fp8_possible = (bool(cfg.general.get("eps_fp8", False))
                        if eps_fp8 is None else bool(eps_fp8)) and \
            self.rollout_mode == "step" and self.steps_per_launch == 1 and \
            self._fp8_layout_ok(self.dims)
        if pair_rollout is not None:
            want_pair = bool(pair_rollout)
        else:
            gate_blocks = 256 if fp8_possible else 384
            want_pair = (bool(cfg.general.get("pair_rollout", False))
                         and self.pairs * self.eps >= gate_blocks)
        self.pair_rollout = (want_pair and self.fused and not self.split_dyn
                             and self.pairs >= 1)
        self._zero_off = torch.zeros(1, dtype=torch.int64, device=d)
        self._zero_sign = torch.zeros(1, dtype=torch.float32, device=d)
        # fp8 sigma*eps stream (halves the dominant HBM bytes AND load count;
        # OCP e4m3fn, exact antithetic cancellation): gated on the pair path,
        # an explicit opt-in, and the row-pair-interleave layout constraints
        # (even input dim + 16-B-aligned weight offsets on vec layers)
        want_fp8 = (bool(cfg.general.get("eps_fp8", False))
                    if eps_fp8 is None else bool(eps_fp8))
        self.eps_fp8 = fp8_possible and self.pair_rollout
        if want_fp8 and not self.eps_fp8 and comm.rank == 0:
            import sys
            print("[engine] eps_fp8 requested but unavailable for this "
                  "model/path — using bf16 eps rows", file=sys.stderr)
        if self.pair_rollout:
            # +1 guard row: the pair forward's prefetch pipeline may overrun
            # the final layer's weights (values are never consumed, but the
            # addresses must stay inside the allocation)
            self._theta_blob = torch.empty((2, self.row_stride), dtype=torch.bfloat16,
                                           device=d)
            self.theta_row = self._theta_blob[:1]
            eps_dt = torch.uint8 if self.eps_fp8 else torch.bfloat16
            self._eps_blob = torch.empty((self.pairs + 1, self.row_stride),
                                         dtype=eps_dt, device=d)
            self.eps_rows = self._eps_blob[:self.pairs]
            self._zeros_n = torch.zeros(self.n, dtype=torch.float32, device=d)
            self._one_signs = torch.ones(self.pairs, dtype=torch.float32, device=d)
        self._warned_host_ranker = False

    @staticmethod
    def _fp8_layout_ok(dims: List[int]) -> bool:
        """Mirror of the fp8 row-pair interleave constraints (pheno.hip):
        each vectorizable layer's PAIR REGION (rows after the plain first
        row of odd input dims) must start 16-byte-aligned so the 16-fp8
        loads stay aligned."""
        off = 0
        for I, O in zip(dims[:-1], dims[1:]):
            vec = (O % 8 == 0) and (off % 8 == 0)
            if vec and (off + (I % 2) * O) % 16 != 0:
                return False
            off += I * O + O
        return True

    def _grad_qstd(self) -> float:
        """Non-zero => the gather re-quantizes each noise value through the
        e4m3 round trip at sigma, so the update uses EXACTLY the perturbation
        values the fp8 rollout evaluated (estimator-exact ES on the quantized
        perturbation distribution)."""
        return float(self.policy.std) if self.eps_fp8 else 0.0

    # ------------------------------------------------------------------ ops
    def _stream(self):
        return torch.cuda.current_stream(self.device).cuda_stream if \
            self.device.type == "cuda" else None

    def _pheno(self):
        std = float(self.policy.std)
        if self.pair_rollout:
            # theta row: sign 0 -> bf16(theta + 0*noise) via the same kernel
            ops.check(ops.hip().es_pheno_bf16(
                self.theta_row.data_ptr(), self.theta.data_ptr(),
                self.nt.noise.data_ptr(), self._zero_off.data_ptr(),
                self._zero_sign.data_ptr(), 1, self.n, self.row_stride, 0.0,
                self._stream()), "es_pheno_bf16")
            if self.eps_fp8:
                # sigma*eps rows in e4m3, row-pair-interleaved (pheno.hip)
                ops.check(ops.hip().es_pheno_fp8(
                    self.eps_rows.data_ptr(), self.nt.noise.data_ptr(),
                    self.offsets.data_ptr(), self.dims_arr.ctypes.data,
                    len(self.dims_arr), self.pairs, self.n, self.row_stride,
                    std, self._stream()), "es_pheno_fp8")
                return
            # sigma*eps rows: zero theta + sign +1 -> bf16(sigma*noise[off_p:])
            ops.check(ops.hip().es_pheno_bf16(
                self.eps_rows.data_ptr(), self._zeros_n.data_ptr(),
                self.nt.noise.data_ptr(), self.offsets.data_ptr(),
                self._one_signs.data_ptr(), self.pairs, self.n, self.row_stride,
                std, self._stream()), "es_pheno_bf16")
            return
        ops.check(ops.hip().es_pheno_bf16(
            self.weights.data_ptr(), self.theta.data_ptr(), self.nt.noise.data_ptr(),
            self.offsets.data_ptr(), self.signs.data_ptr(), self.M, self.n,
            self.row_stride, std, self._stream()), "es_pheno_bf16")

    def _forward(self, obs: torch.Tensor, salt: int):
        ops.check(ops.hip().es_mlp_fwd(
            self.actions.data_ptr(), obs.data_ptr(), self.weights.data_ptr(),
            self.obmean.data_ptr(), self.obstd.data_ptr(),
            self.dims_arr.ctypes.data, len(self.dims_arr), self.seed_dev.data_ptr(),
            salt, self.B, float(self.policy._module.ob_clip), self.acstd_dev.data_ptr(),
            self.row_stride, 1, (self.M - 1) * self.eps, self.bins, self.eps,
            self.act_mode,
            self.alow_dev.data_ptr() if self.alow_dev is not None else None,
            self.arange_dev.data_ptr() if self.arange_dev is not None else None,
            self._stream()), "es_mlp_fwd")
        return self.actions

    def _loco_step(self, t: int):
        """One fused rollout step for the 2*pairs perturbed members; the
        noiseless member runs as a side-stream whole-episode kernel so the
        main grid stays an exact multiple of the CU slot count."""
        env = self.env
        goal_ptr = env.goal.data_ptr() if env.goal_conditioned else None
        if self.pair_rollout:
            fn = ops.hip().es_loco_pair_step_fp8 if self.eps_fp8 else \
                ops.hip().es_loco_pair_step
            ops.check(fn(
                self.theta_row.data_ptr(), self.eps_rows.data_ptr(),
                self.obmean.data_ptr(), self.obstd.data_ptr(),
                self.dims_arr.ctypes.data, len(self.dims_arr),
                self.seed_dev.data_ptr(), t + 1,
                float(self.policy._module.ob_clip), self.acstd_dev.data_ptr(),
                self.row_stride,
                env.s.data_ptr(), env.pos.data_ptr(), goal_ptr,
                env.A_bf16.data_ptr(), env.B.data_ptr(), env.b0.data_ptr(),
                env.wv.data_ptr(), env.wa.data_ptr(), env.wy.data_ptr(),
                env.wh.data_ptr(),
                self.alive.data_ptr(), self.rew_total.data_ptr(),
                self.member_steps.data_ptr(), self.behv.data_ptr(),
                self.mo_sum.data_ptr(), self.mo_sumsq.data_ptr(),
                self.pairs, env.sdim, env.ac_dim, int(env.goal_conditioned),
                int(env.terminate_on_fall), (self.M - 1) * self.eps, self.bins,
                self.eps, self.act_mode,
                float(env.leak), float(env.ctrl_cost), float(env.alive_bonus),
                float(env.fall_threshold), float(env.dt), self._stream()),
                "es_loco_pair_step_fp8" if self.eps_fp8 else "es_loco_pair_step")
            return
        common = (
            self.weights.data_ptr(), self.obmean.data_ptr(), self.obstd.data_ptr(),
            self.dims_arr.ctypes.data, len(self.dims_arr), self.seed_dev.data_ptr(),
            t + 1, float(self.policy._module.ob_clip), self.acstd_dev.data_ptr(),
            self.row_stride,
            env.s.data_ptr(), env.pos.data_ptr(), goal_ptr,
            env.A_bf16.data_ptr(), env.B.data_ptr(), env.b0.data_ptr(),
            env.wv.data_ptr(), env.wa.data_ptr(), env.wy.data_ptr(), env.wh.data_ptr(),
            self.alive.data_ptr(), self.rew_total.data_ptr(),
            self.member_steps.data_ptr(), self.behv.data_ptr(),
            self.mo_sum.data_ptr(), self.mo_sumsq.data_ptr(),
            (self.M - 1) * self.eps, env.sdim, env.ac_dim, int(env.goal_conditioned),
            int(env.terminate_on_fall), (self.M - 1) * self.eps, self.bins, self.eps,
            self.act_mode,
            float(env.leak), float(env.ctrl_cost), float(env.alive_bonus),
            float(env.fall_threshold), float(env.dt))
        if self.split_dyn:
            ops.check(ops.hip().es_loco_step_split(
                *common, self.act_scratch.data_ptr(), self.dyn_group,
                self._stream()), "es_loco_step_split")
        else:
            ops.check(ops.hip().es_loco_step(*common, self._stream()), "es_loco_step")

    def _loco_episode(self, member_base: int, n_members: int, noiseless_from: int,
                      n_steps: Optional[int] = None, salt_base: int = 0,
                      weights_ptr: Optional[int] = None, wrow0: int = 0,
                      block_threads: int = 256):
        """n_steps consecutive env steps (default: the whole episode) for
        [member_base, member_base+n_members) slots in one launch. weights_ptr
        + wrow0 let the call run on a sub-blob (pair mode: the 1-row theta
        blob serves the noiseless slots, whose weights row is wrow0)."""
        env = self.env
        goal_ptr = env.goal.data_ptr() if env.goal_conditioned else None
        ops.check(ops.hip().es_loco_episode(
            self.weights.data_ptr() if weights_ptr is None else weights_ptr,
            self.obmean.data_ptr(), self.obstd.data_ptr(),
            self.dims_arr.ctypes.data, len(self.dims_arr), self.seed_dev.data_ptr(),
            self.max_steps if n_steps is None else n_steps,
            float(self.policy._module.ob_clip),
            self.acstd_dev.data_ptr(), self.row_stride,
            env.s.data_ptr(), env.pos.data_ptr(), goal_ptr,
            env.A_bf16.data_ptr(), env.B.data_ptr(), env.b0.data_ptr(),
            env.wv.data_ptr(), env.wa.data_ptr(), env.wy.data_ptr(), env.wh.data_ptr(),
            self.alive.data_ptr(), self.rew_total.data_ptr(),
            self.member_steps.data_ptr(), self.behv.data_ptr(),
            self.mo_sum.data_ptr(), self.mo_sumsq.data_ptr(),
            member_base, n_members, salt_base, wrow0, env.sdim, env.ac_dim,
            int(env.goal_conditioned), int(env.terminate_on_fall), noiseless_from,
            self.bins, self.eps, self.act_mode,
            float(env.leak), float(env.ctrl_cost), float(env.alive_bonus),
            float(env.fall_threshold), float(env.dt), int(block_threads),
            self._stream()), "es_loco_episode")

    def _loco_pair_episode(self, n_steps: Optional[int] = None, salt_base: int = 0):
        """n_steps consecutive env steps for every (pair, episode) block in
        ONE launch (default: the whole episode). Blocks never rendezvous, so
        dynamics-L2 phases overlap other blocks' HBM weight streaming —
        the per-step pair grid measured only ~47% HBM duty (phase-locked
        launches); trajectories are bitwise-identical to per-step launches."""
        env = self.env
        goal_ptr = env.goal.data_ptr() if env.goal_conditioned else None
        ops.check(ops.hip().es_loco_pair_episode(
            self.theta_row.data_ptr(), self.eps_rows.data_ptr(),
            self.obmean.data_ptr(), self.obstd.data_ptr(),
            self.dims_arr.ctypes.data, len(self.dims_arr),
            self.seed_dev.data_ptr(),
            float(self.policy._module.ob_clip), self.acstd_dev.data_ptr(),
            self.row_stride,
            env.s.data_ptr(), env.pos.data_ptr(), goal_ptr,
            env.A_bf16.data_ptr(), env.B.data_ptr(), env.b0.data_ptr(),
            env.wv.data_ptr(), env.wa.data_ptr(), env.wy.data_ptr(),
            env.wh.data_ptr(),
            self.alive.data_ptr(), self.rew_total.data_ptr(),
            self.member_steps.data_ptr(), self.behv.data_ptr(),
            self.mo_sum.data_ptr(), self.mo_sumsq.data_ptr(),
            self.pairs, env.sdim, env.ac_dim, int(env.goal_conditioned),
            int(env.terminate_on_fall), (self.M - 1) * self.eps, self.bins,
            self.eps, self.act_mode,
            float(env.leak), float(env.ctrl_cost), float(env.alive_bonus),
            float(env.fall_threshold), float(env.dt),
            self.max_steps if n_steps is None else int(n_steps), int(salt_base),
            self._stream()), "es_loco_pair_episode")

    def _loco_noiseless_episode(self):
        # 512-thread blocks: the single-member episode is latency-serial, so
        # doubling PART halves its dependency chains (the population grids
        # keep 256-thread blocks for occupancy)
        if self.pair_rollout:
            self._loco_episode((self.M - 1) * self.eps, self.eps, 0,
                               weights_ptr=self.theta_row.data_ptr(),
                               wrow0=self.M - 1, block_threads=512)
        else:
            self._loco_episode((self.M - 1) * self.eps, self.eps, 0,
                               block_threads=512)

    # ------------------------------------------------------------- rollout
    def _step_body(self, t: int):
        a = self._forward(self.obs_buf, salt=t + 1)
        ob, rew, done = self.env.step(a)
        alive = self.alive
        self.rew_total.add_(rew * alive)
        self.member_steps.add_(alive)
        am = alive.bool().unsqueeze(1)
        self.behv.copy_(torch.where(am, self.env.positions, self.behv))
        # per-gen observation statistics, save_mask-gated like the reference's
        # save_obs_chance episodes (simple_example.py:38, obstat fed at es.py:73-74)
        w = alive * self.save_mask
        self.ob_sum.add_((ob * w.unsqueeze(1)).sum(0).double())
        self.ob_sumsq.add_((ob * ob * w.unsqueeze(1)).sum(0).double())
        self.ob_count.add_(w.sum().double())
        self.alive.mul_(1.0 - done.float())
        self.obs_buf.copy_(ob)

    def _rollout_body(self):
        if self.fused and self.rollout_mode == "episode":
            # whole generation in one (+1 side) launch: every block runs its
            # member's full episode, blocks drift freely (no per-step
            # rendezvous); noiseless member on the side stream keeps the main
            # grid an exact multiple of the CU slot count
            main = torch.cuda.current_stream(self.device)
            self._side.wait_stream(main)
            with torch.cuda.stream(self._side):
                self._loco_noiseless_episode()
            if self.pair_rollout:
                self._loco_pair_episode()
            else:
                self._loco_episode(0, (self.M - 1) * self.eps, (self.M - 1) * self.eps)
            main.wait_stream(self._side)
            return
        if self.fused:
            # noiseless member: one whole-episode kernel on a side stream,
            # concurrent with the per-step population kernels
            main = torch.cuda.current_stream(self.device)
            self._side.wait_stream(main)
            with torch.cuda.stream(self._side):
                self._loco_noiseless_episode()
            k = self.steps_per_launch
            if k > 1:
                mm = (self.M - 1) * self.eps
                for t0 in range(0, self.max_steps, k):
                    n = min(k, self.max_steps - t0)
                    if self.pair_rollout:
                        self._loco_pair_episode(n_steps=n, salt_base=t0)
                    else:
                        self._loco_episode(0, mm, mm, n_steps=n, salt_base=t0)
            else:
                for t in range(self.max_steps):
                    self._loco_step(t)
            main.wait_stream(self._side)
        else:
            for t in range(self.max_steps):
                self._step_body(t)

    def _rollout(self):
        body = self._loco_step if self.fused else self._step_body
        if self.use_graph:
            if self._graph is None:
                # warmup (lazy inits must happen outside capture), then capture once
                s = torch.cuda.Stream(self.device)
                s.wait_stream(torch.cuda.current_stream(self.device))
                with torch.cuda.stream(s):
                    for t in range(3):
                        body(t)
                    if self.fused:
                        self._loco_noiseless_episode()  # warm the episode kernel
                torch.cuda.current_stream(self.device).wait_stream(s)
                # warmup dirtied the rollout state: restore it before capture
                self._reset_rollout_state()
                self.obs_buf.copy_(self.env.reset(self._gen_seed()))
                self._graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self._graph):
                    self._rollout_body()
            self._graph.replay()
            return
        self._rollout_body()

    def _reset_rollout_state(self):
        self.alive.fill_(1.0)
        self.rew_total.zero_()
        self.member_steps.zero_()
        self.behv.zero_()
        self.ob_sum.zero_()
        self.ob_sumsq.zero_()
        self.ob_count.zero_()
        if self.fused:
            self.mo_sum.zero_()
            self.mo_sumsq.zero_()

    def _gen_seed(self) -> int:
        # distinct env variations per generation, identical across ranks only
        # for the matrices (seeded in env ctor); initial states vary per rank
        return (self.gen * 1000003 + self.comm.rank * 7919) & 0x7FFFFFFF

    # ------------------------------------------------------------- fitness
    def _member_rewards(self) -> torch.Tensor:
        """(M,) episode-averaged total reward per member (obj.py:56-63)."""
        return self.rew_total.view(self.M, self.eps).mean(1)

    def _member_behv(self) -> torch.Tensor:
        """(M, 3) final behaviour; last episode's, like the reference's
        multi-episode rollout (flagrun.py:105-140 keeps the last behv)."""
        return self.behv.view(self.M, self.eps, 3)[:, -1]

    def _fitnesses(self) -> torch.Tensor:
        """(M, O) fitness per member from accumulated rollout state."""
        rew = self._member_rewards()
        behv = self._member_behv()
        if self.objective == "reward":
            return rew.unsqueeze(1)
        if self.objective == "mean_reward":
            steps = self.member_steps.view(self.M, self.eps).mean(1)
            return (rew / torch.clamp(steps, min=1.0)).unsqueeze(1)
        if self.objective == "dist":
            return behv[:, :2].norm(dim=1, keepdim=True)
        if self.objective == "xdist":
            return behv[:, :1]
        if self.objective in ("ns", "nsr"):
            assert self.archive is not None, "set engine.archive before ns/nsr generations"
            nov = novelty_batch(behv[:, :2], self.archive, self.novelty_k)
            if self.objective == "ns":
                return nov.unsqueeze(1)
            return torch.stack([rew, nov], dim=1)
        raise ValueError(f"unknown objective {self.objective!r}")

    def noiseless_eval(self) -> Tuple[float, np.ndarray, int]:
        """Evaluation-only noiseless episode: (reward, behaviour(3,), steps).

        Runs ONLY the unperturbed policy — no optimizer update, no numpy RNG
        draws, nothing mutated but the rollout scratch buffers. Mirrors the
        reference's archive-init evaluation (``nsra.py:31-45``: noiseless
        rollouts with no training step).
        """
        if self.pair_rollout:
            ops.check(ops.hip().es_pheno_bf16(
                self.theta_row.data_ptr(), self.theta.data_ptr(),
                self.nt.noise.data_ptr(), self._zero_off.data_ptr(),
                self._zero_sign.data_ptr(), 1, self.n, self.row_stride, 0.0,
                self._stream()), "es_pheno_bf16")
        elif self.device.type == "cuda":
            # bf16(theta) into the noiseless slot's weights row (sign 0)
            ops.check(ops.hip().es_pheno_bf16(
                self.weights[self.M - 1:].data_ptr(), self.theta.data_ptr(),
                self.nt.noise.data_ptr(), self._zero_off.data_ptr(),
                self._zero_sign.data_ptr(), 1, self.n, self.row_stride, 0.0,
                self._stream()), "es_pheno_bf16")
        else:
            self.weights[self.M - 1, :self.n] = self.theta.bfloat16()
        self._reset_rollout_state()
        self.obs_buf.copy_(self.env.reset(self._gen_seed()))
        self.acstd_dev.fill_(0.0)  # reference noiseless eval: use_ac_noise False
        if self.fused:
            self._loco_noiseless_episode()
        else:
            # generic torch path computes the whole batch; only the noiseless
            # slots are read out (init-time only, cost irrelevant)
            self.weights.copy_(self.weights[self.M - 1:].expand_as(self.weights))
            for t in range(self.max_steps):
                self._step_body(t)
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        rew = float(self._member_rewards()[-1].item())
        behv = self._member_behv()[-1].cpu().numpy()
        steps = int(self.member_steps[(self.M - 1) * self.eps:].sum().item())
        return rew, behv, steps

    # ---------------------------------------------------------------- step
    def step(self, ranker: Ranker, reporter=None) -> Tuple[_NoiselessResult, ObStat]:
        """Run one full generation; mirrors ``es.step`` semantics (``es.py:23-51``)."""
        t0 = time.perf_counter()
        cfg = self.cfg
        if self._offs_prefetched:
            # pheno (main stream) must see the side-stream offset upload
            torch.cuda.current_stream(self.device).wait_stream(self._side)
        else:
            self._upload_offsets()
        self._offs_prefetched = False
        self.seed_dev.fill_(int(self.rs.randint(0, 2 ** 31)))
        self.acstd_dev.fill_(float(getattr(self.policy._module, "_action_std", 0.0)))
        chance = float(cfg.policy.get("save_obs_chance", 1.0))
        sm = (self.rs.random_sample(self.B) < chance).astype(np.float32)
        # noiseless slots never feed the gen obstat (reference es.py:48 is separate)
        sm[(self.M - 1) * self.eps:] = 0.0
        self.save_mask.copy_(torch.from_numpy(sm).to(self.device))

        self._pheno()
        self._reset_rollout_state()
        self.obs_buf.copy_(self.env.reset(self._gen_seed()))
        t1 = time.perf_counter()
        self._rollout()
        fits = self._fitnesses()

        # -- share (fit+, fit-, idx) triples: ONE RCCL all_gather over xGMI
        O = fits.shape[1]
        rows = torch.empty((self.pairs, 2 * O + 1), dtype=torch.float64, device=self.device)
        rows[:, :O] = fits[:self.pairs].double()
        rows[:, O:2 * O] = fits[self.pairs:2 * self.pairs].double()
        rows[:, -1] = self.offsets[:self.pairs].double()
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t2 = time.perf_counter()
        # overlap: this generation's offsets are already captured in `rows`,
        # so the NEXT generation's noise draw uploads on a side stream while
        # the fitness all-gather is in flight over xGMI
        if self._side is not None:
            with torch.cuda.stream(self._side):
                self._upload_offsets()
            self._offs_prefetched = True
        all_rows_dev = self.comm.allgather_rows(rows)
        # evaluated-episode steps only (the reference's es.py:79 counts the
        # pos+neg rollouts, not the noiseless eval)
        local_steps = float(self.member_steps[:2 * self.pairs * self.eps].sum().item())
        steps = int(self.comm.allreduce_scalar(local_steps))
        t3 = time.perf_counter()

        # -- identical redundant ranking on every rank. K4 device path: a
        # plain single-objective CenteredRanker runs as torch argsort on
        # device (stable sort: deterministic, rank-identical), so the
        # gradient + Adam launch immediately and the host mirror copy for
        # reporters/heuristics overlaps them. Other rankers take the host
        # path (pop-sized, sub-ms).
        dev_rank = (type(ranker) is CenteredRanker and O == 1
                    and self.device.type == "cuda"
                    and all_rows_dev.device.type == "cuda")
        if dev_rank:
            fits_all = torch.cat([all_rows_dev[:, 0], all_rows_dev[:, 1]]).float()
            npop = fits_all.numel()
            r = torch.empty(npop, dtype=torch.float32, device=self.device)
            r[torch.argsort(fits_all, stable=True)] = torch.arange(
                npop, dtype=torch.float32, device=self.device)
            y = r / (npop - 1) - 0.5
            rf = (y[:npop // 2] - y[npop // 2:]).contiguous()
            ri = all_rows_dev[:, 2].long().contiguous()
            ops.check(ops.hip().es_grad_gather(
                self.grad.data_ptr(), self.nt.noise.data_ptr(), rf.data_ptr(),
                ri.data_ptr(), rf.numel(), self.n, self._grad_qstd(),
                self._stream()), "es_grad_gather")
            self._optim_step(float(npop), float(cfg.policy.l2coeff))
            # host mirror (for reporters and the entry scripts' heuristics)
            # overlaps the update kernels above
            all_rows = all_rows_dev.cpu().numpy()
            ranker._pre_rank(all_rows[:, :O], all_rows[:, O:2 * O], all_rows[:, -1])
            ranker.ranked_fits = rf.cpu().numpy().astype(np.float64)
            ranker.n_fits_ranked = npop
        else:
            if not self._warned_host_ranker and self.device.type == "cuda":
                self._warned_host_ranker = True
                if type(ranker) is not CenteredRanker and self.comm.rank == 0:
                    import sys
                    print(f"[engine] ranking on host: {type(ranker).__name__} "
                          f"(device fast path covers plain single-objective "
                          f"CenteredRanker only; pop-sized, sub-ms)",
                          file=sys.stderr)
            all_rows = all_rows_dev.cpu().numpy()
            pos, neg = all_rows[:, :O], all_rows[:, O:2 * O]
            inds = all_rows[:, -1]
            ranker.rank(pos, neg, inds)
            rf = torch.from_numpy(np.ascontiguousarray(ranker.ranked_fits,
                                                       dtype=np.float32)).to(self.device)
            ri = torch.from_numpy(np.ascontiguousarray(ranker.noise_inds,
                                                       dtype=np.int64)).to(self.device)
            ops.check(ops.hip().es_grad_gather(
                self.grad.data_ptr(), self.nt.noise.data_ptr(), rf.data_ptr(),
                ri.data_ptr(), rf.numel(), self.n, self._grad_qstd(),
                self._stream()), "es_grad_gather")
            self._optim_step(float(ranker.n_fits_ranked), float(cfg.policy.l2coeff))
        self.sync_host(light=True)
        pos, neg = all_rows[:, :O], all_rows[:, O:2 * O]
        t4 = time.perf_counter()

        # -- per-gen obstat -> merged across ranks (packed all_reduce)
        gen_obstat = ObStat(self.env.observation_space.shape, 0)
        if self.fused:
            # fused kernel accumulated per-member alive-weighted sums; apply
            # the save_obs_chance member flags at reduction time
            sm = self.save_mask.unsqueeze(1)
            ob_sum = (self.mo_sum * sm).sum(0).double().cpu().numpy()
            ob_sumsq = (self.mo_sumsq * sm).sum(0).double().cpu().numpy()
            ob_count = float((self.member_steps * self.save_mask).sum().item())
            gen_obstat.inc(ob_sum, ob_sumsq, ob_count)
        else:
            gen_obstat.inc(self.ob_sum.cpu().numpy(), self.ob_sumsq.cpu().numpy(),
                           float(self.ob_count.item()))
        gen_obstat.dist_inc(self.comm)

        nl_rew = float(self._member_rewards()[-1].item())
        nl_pos = self._member_behv()[-1].cpu().numpy()
        noiseless = _NoiselessResult(nl_rew, nl_pos, steps, self.env.observation_space.shape)
        if reporter is not None:
            reporter.log_gen(np.concatenate([pos, neg]), noiseless, self.policy, steps)

        self.gen += 1
        self.timings = {"pheno_s": t1 - t0, "rollout_s": t2 - t1, "comm_s": t3 - t2,
                        "update_s": t4 - t3, "gen_s": time.perf_counter() - t0,
                        "env_steps": steps}
        return noiseless, gen_obstat

    def grow_archive(self) -> float:
        """Device-resident NSR-A archive growth (reference ``nsra.py:130-135``
        + ``novelty.py:9-13``): append rank-0's noiseless behaviour to the
        device archive (ONE 2-float RCCL broadcast — the entry must be
        identical on every rank) and return its novelty against the archive
        it joined. Call once per generation after :meth:`step`; stays on
        device end to end."""
        assert self.archive is not None, "set engine.archive first"
        b = self._member_behv()[-1][:2].double().contiguous()
        self.comm.broadcast_tensor_(b, src=0)
        nov = float(novelty_batch(b.unsqueeze(0), self.archive,
                                  self.novelty_k)[0].item())
        self.archive = torch.cat([self.archive, b.unsqueeze(0)])
        return nov

    def _upload_offsets(self):
        """Sample this rank's antithetic noise offsets and upload to device."""
        offs = self.nt.sample_idxs(self.rs, self.pairs)
        self.offsets[:self.pairs].copy_(torch.from_numpy(offs).to(self.device,
                                                                  non_blocking=True))
        self.offsets[self.pairs:2 * self.pairs].copy_(self.offsets[:self.pairs])

    # ------------------------------------------------------------- update
    def _optim_step(self, n_ranked: float, l2: float):
        opt = self.policy.optim
        opt.t += 1
        gscale = 1.0 / n_ranked
        if isinstance(opt, Adam):
            a = opt.lr * np.sqrt(1 - opt.beta2 ** opt.t) / (1 - opt.beta1 ** opt.t)
            ops.check(ops.hip().es_adam_step(
                self.theta.data_ptr(), self.m.data_ptr(), self.v.data_ptr(),
                self.grad.data_ptr(), self.n, float(a), float(opt.beta1), float(opt.beta2),
                float(opt.epsilon), l2, gscale, self._stream()), "es_adam_step")
        elif isinstance(opt, SGD):
            ops.check(ops.hip().es_sgd_step(
                self.theta.data_ptr(), self.v.data_ptr(), self.grad.data_ptr(), self.n,
                float(opt.lr), float(opt.momentum), l2, gscale, self._stream()),
                "es_sgd_step")
        else:
            raise TypeError(f"GpuEngine supports Adam/SGD, got {type(opt).__name__}")

    def _load_optim_state(self):
        opt: Optimizer = self.policy.optim
        if isinstance(opt, Adam) and np.any(opt.m):
            self.m.copy_(torch.from_numpy(opt.m).to(self.device)[self.perm])
            self.v.copy_(torch.from_numpy(opt.v).to(self.device)[self.perm])
        elif isinstance(opt, SGD) and np.any(opt.v):
            self.v.copy_(torch.from_numpy(opt.v).to(self.device)[self.perm])

    def _push_obstat(self):
        self.obmean.copy_(torch.from_numpy(
            np.asarray(self.policy.obstat.mean, dtype=np.float32)).to(self.device))
        self.obstd.copy_(torch.from_numpy(
            np.asarray(self.policy.obstat.std, dtype=np.float32)).to(self.device))

    def update_obstat(self, gen_obstat: ObStat):
        """Fold the generation's stats into the lifetime stat (reference
        ``policy.update_obstat``, ``policy.py:69-71``) and push to device."""
        self.policy.update_obstat(gen_obstat)
        self._push_obstat()

    def noise_slice_flat(self, idx: int) -> np.ndarray:
        """Noise slice at ``idx`` permuted into the FLAT state_dict layout.

        On the engine path noise element t perturbs forward-layout parameter
        t, so reconstructing a perturbed individual for ``Policy.pheno``
        (flat layout) needs the inverse permutation.
        """
        fwd = self.nt.get(int(idx), self.n)
        flat = torch.empty_like(fwd)
        flat[self.perm] = fwd
        return flat.cpu().numpy()

    def sync_host(self, light: bool = False):
        """Mirror device truth into the host Policy (checkpoint compatibility).

        light=True syncs flat_params only (once per generation); full sync
        also writes optimizer moments so ``Policy.save`` emits the exact
        reference pickle format with optimizer state included.
        """
        flat = torch.empty(self.n, dtype=torch.float32, device=self.device)
        flat[self.perm] = self.theta
        self.policy.flat_params[:] = flat.cpu().numpy()
        # failure detection: a diverged update (NaN/inf params) must stop the
        # run loudly instead of silently training garbage for hours
        if not np.isfinite(self.policy.flat_params).all():
            raise FloatingPointError(
                f"non-finite parameters after generation {self.gen} "
                f"(lr={self.policy.optim.lr}, std={self.policy.std}) — "
                "lower lr/noise std or check fitness scaling")
        if not light:
            opt = self.policy.optim
            inv = torch.empty_like(flat)
            if isinstance(opt, Adam):
                inv[self.perm] = self.m
                opt.m[:] = inv.cpu().numpy()
                inv[self.perm] = self.v
                opt.v[:] = inv.cpu().numpy()
            elif isinstance(opt, SGD):
                inv[self.perm] = self.v
                opt.v[:] = inv.cpu().numpy()
            self.policy.set_nn_params(self.policy.flat_params)

    def restore_from_policy(self, gen: Optional[int] = None):
        """Re-push host Policy state into the device buffers after an
        in-place checkpoint restore (utils/checkpoint.py): params, optimizer
        moments, obs normalization, and optionally the generation counter
        (which seeds per-gen env variations, so exact resume needs it)."""
        flat = torch.from_numpy(self.policy.flat_params).to(self.device)
        self.theta.copy_(flat[self.perm])
        self.m.zero_()
        self.v.zero_()
        self._load_optim_state()
        self._push_obstat()
        self._offs_prefetched = False  # pre-restore offset draws are stale
        if gen is not None:
            self.gen = int(gen)

    def checkpoint_state(self) -> dict:
        """Per-rank engine internals for RunCheckpointer beyond the Policy:
        the generation counter and the side-stream-prefetched next-gen noise
        offsets. The prefetch consumes rs draws at the END of a step, so a
        resume that re-drew them would desync from the uninterrupted RNG
        stream — the drawn values must travel with the snapshot."""
        st = {"gen": self.gen, "prefetched": None}
        if self._offs_prefetched:
            if self._side is not None:
                torch.cuda.current_stream(self.device).wait_stream(self._side)
            st["prefetched"] = self.offsets[:self.pairs].cpu().numpy()
        return st

    def load_checkpoint_state(self, st: dict):
        self.gen = int(st["gen"])
        if st.get("prefetched") is not None:
            offs = torch.from_numpy(np.asarray(st["prefetched"])).to(self.device)
            self.offsets[:self.pairs].copy_(offs)
            self.offsets[self.pairs:2 * self.pairs].copy_(offs)
            self._offs_prefetched = True
