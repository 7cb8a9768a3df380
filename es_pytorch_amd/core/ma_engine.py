"""MultiAgentGpuEngine — GPU-batched co-evolution (beyond reference parity).

The reference evaluates co-evolving policies in ONE Unity sim per CPU rank,
sequentially (``multi_agent.py:33-67``). Here N policies' perturbations play
against each other in B batched env instances per generation: env instance b
is a joint game between perturbation b of EVERY policy. Per env step each
policy's population forward runs as one HIP kernel (``mlp_fwd.hip``) over
its agents' observations, the batched multi-agent env consumes the action
list, and per-agent rewards accumulate under a shared alive mask.

Per-policy updates mirror the single-policy engine: RCCL all_gather of that
policy's (fit+, fit-, idx) triples, redundant host ranking, gather-GEMV
gradient from the shared noise table, fused Adam — all per policy, from its
own fitness column (reference ``multi_agent.py:110-125``).

Unlike the reference (which evaluates the SAME +noise phenotypes twice —
``multi_agent.py:48-49``, a documented quirk), this engine uses true
antithetic pairs: instance slots [0, pairs) carry +noise for every policy
and [pairs, 2*pairs) carry -noise; slot 2*pairs is the all-noiseless game.
"""
from __future__ import annotations

import time
from typing import List, Tuple

import numpy as np
import torch

from es_pytorch_amd import ops
from es_pytorch_amd.core.engine import forward_perm
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.utils.rankers import Ranker


class _PolicyState:
    """Per-policy device state (theta, moments, member weight blobs)."""

    def __init__(self, policy: Policy, device, pairs: int, ob_dim: int):
        self.policy = policy
        d = device
        self.dims = policy._module.layer_dims()
        self.n = len(policy)
        self.perm = forward_perm(self.dims).to(d)
        self.dims_arr = np.array(self.dims, dtype=np.int32)
        flat = torch.from_numpy(policy.flat_params).to(d)
        self.theta = flat[self.perm].contiguous()
        self.m = torch.zeros(self.n, dtype=torch.float32, device=d)
        self.v = torch.zeros(self.n, dtype=torch.float32, device=d)
        self.row_stride = (self.n + 7) // 8 * 8
        M = 2 * pairs + 1
        self.weights = torch.empty((M, self.row_stride), dtype=torch.bfloat16, device=d)
        self.offsets = torch.zeros(M, dtype=torch.int64, device=d)
        self.signs = torch.cat([torch.ones(pairs), -torch.ones(pairs),
                                torch.zeros(1)]).to(d)
        self.acstd_dev = torch.zeros(1, dtype=torch.float32, device=d)
        self.grad = torch.empty(self.n, dtype=torch.float32, device=d)
        self.obmean = torch.zeros(ob_dim, dtype=torch.float32, device=d)
        self.obstd = torch.ones(ob_dim, dtype=torch.float32, device=d)
        self.push_obstat()

    def push_obstat(self):
        self.obmean.copy_(torch.from_numpy(
            np.asarray(self.policy.obstat.mean, dtype=np.float32)))
        self.obstd.copy_(torch.from_numpy(
            np.asarray(self.policy.obstat.std, dtype=np.float32)))

    def sync_host(self):
        flat = torch.empty(self.n, dtype=torch.float32, device=self.theta.device)
        flat[self.perm] = self.theta
        self.policy.flat_params[:] = flat.cpu().numpy()
        if not np.isfinite(self.policy.flat_params).all():
            raise FloatingPointError("non-finite parameters in co-evolution update")


class MultiAgentGpuEngine:
    def __init__(self, cfg, comm: Comm, policies: List[Policy], nt: NoiseTable, env,
                 rs: np.random.RandomState):
        self.cfg = cfg
        self.comm = comm
        self.nt = nt
        self.env = env
        self.rs = rs
        self.device = nt.noise.device
        self.n_agents = len(policies)
        assert env.N_AGENTS == self.n_agents

        ppg = cfg.general.policies_per_gen
        assert ppg % comm.size == 0 and (ppg / comm.size) % 2 == 0
        self.pairs = int(ppg // comm.size // 2)
        self.B = 2 * self.pairs + 1
        assert env.batch == self.B, f"env batch {env.batch} != {self.B}"

        self.states = [_PolicyState(p, self.device, self.pairs, env.ob_dims[i])
                       for i, p in enumerate(policies)]
        self.actions = [torch.empty((self.B, env.ac_dims[i]), dtype=torch.float32,
                                    device=self.device) for i in range(self.n_agents)]
        self.alive = torch.ones(self.B, dtype=torch.float32, device=self.device)
        self.rew_total = torch.zeros((self.B, self.n_agents), dtype=torch.float32,
                                     device=self.device)
        self.member_steps = torch.zeros(self.B, dtype=torch.float32, device=self.device)
        self.seed_dev = torch.zeros(1, dtype=torch.int64, device=self.device)
        self.max_steps = int(cfg.env.max_steps)
        self.gen = 0
        self.timings = {}

    def _stream(self):
        return torch.cuda.current_stream(self.device).cuda_stream if \
            self.device.type == "cuda" else None

    def _forward(self, i: int, obs: torch.Tensor, salt: int):
        st = self.states[i]
        ops.check(ops.hip().es_mlp_fwd(
            self.actions[i].data_ptr(), obs.contiguous().data_ptr(),
            st.weights.data_ptr(), st.obmean.data_ptr(), st.obstd.data_ptr(),
            st.dims_arr.ctypes.data, len(st.dims_arr), self.seed_dev.data_ptr(),
            salt * self.n_agents + i, self.B,
            float(st.policy._module.ob_clip), st.acstd_dev.data_ptr(),
            st.row_stride, 1, self.B - 1, 0, 1, 0, None, None,
            self._stream()), "es_mlp_fwd")
        return self.actions[i]

    def step(self, rankers: List[Ranker]) -> Tuple[List[float], List[ObStat]]:
        """One co-evolution generation; every policy updated from its own
        fitness column. :returns: (noiseless rewards per agent, obstats)."""
        t0 = time.perf_counter()
        for i, st in enumerate(self.states):
            offs = self.nt.sample_idxs(self.rs, self.pairs)  # one draw per policy
            st.offsets[:self.pairs].copy_(torch.from_numpy(offs).to(self.device))
            st.offsets[self.pairs:2 * self.pairs].copy_(st.offsets[:self.pairs])
            st.acstd_dev.fill_(float(getattr(st.policy._module, "_action_std", 0.0)))
            ops.check(ops.hip().es_pheno_bf16(
                st.weights.data_ptr(), st.theta.data_ptr(), self.nt.noise.data_ptr(),
                st.offsets.data_ptr(), st.signs.data_ptr(), self.B, st.n,
                st.row_stride, float(st.policy.std), self._stream()), "es_pheno_bf16")
        self.seed_dev.fill_(int(self.rs.randint(0, 2 ** 31)))

        self.alive.fill_(1.0)
        self.rew_total.zero_()
        self.member_steps.zero_()
        obs = self.env.reset((self.gen * 1000003 + self.comm.rank * 7919) & 0x7FFFFFFF)
        ob_sums = [torch.zeros(self.env.ob_dims[i], dtype=torch.float64,
                               device=self.device) for i in range(self.n_agents)]
        ob_sumsqs = [torch.zeros_like(s) for s in ob_sums]
        ob_count = torch.zeros((), dtype=torch.float64, device=self.device)

        for t in range(self.max_steps):
            acts = [self._forward(i, obs[i], t + 1) for i in range(self.n_agents)]
            obs, rews, done = self.env.step(acts)
            self.rew_total.add_(rews * self.alive.unsqueeze(1))
            self.member_steps.add_(self.alive)
            w = self.alive.unsqueeze(1)
            for i in range(self.n_agents):
                ob_sums[i].add_((obs[i] * w).sum(0).double())
                ob_sumsqs[i].add_((obs[i] * obs[i] * w).sum(0).double())
            ob_count.add_(self.alive.sum().double())
            self.alive.mul_(1.0 - done.float())

        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t1 = time.perf_counter()

        # per-policy: share triples, rank, reconstruct gradient, update
        noiseless = []
        obstats = []
        for i, st in enumerate(self.states):
            fits = self.rew_total[:, i]
            rows = torch.empty((self.pairs, 3), dtype=torch.float64, device=self.device)
            rows[:, 0] = fits[:self.pairs].double()
            rows[:, 1] = fits[self.pairs:2 * self.pairs].double()
            rows[:, 2] = st.offsets[:self.pairs].double()
            all_rows = self.comm.allgather_rows(rows).cpu().numpy()
            rankers[i].rank(all_rows[:, :1], all_rows[:, 1:2], all_rows[:, 2])
            rf = torch.from_numpy(np.ascontiguousarray(
                rankers[i].ranked_fits, dtype=np.float32)).to(self.device)
            ri = torch.from_numpy(np.ascontiguousarray(
                rankers[i].noise_inds, dtype=np.int64)).to(self.device)
            ops.check(ops.hip().es_grad_gather(
                st.grad.data_ptr(), self.nt.noise.data_ptr(), rf.data_ptr(),
                ri.data_ptr(), rf.numel(), st.n, 0.0, self._stream()), "es_grad_gather")
            opt = st.policy.optim
            assert isinstance(opt, Adam)
            opt.t += 1
            a = opt.lr * np.sqrt(1 - opt.beta2 ** opt.t) / (1 - opt.beta1 ** opt.t)
            ops.check(ops.hip().es_adam_step(
                st.theta.data_ptr(), st.m.data_ptr(), st.v.data_ptr(),
                st.grad.data_ptr(), st.n, float(a), float(opt.beta1), float(opt.beta2),
                float(opt.epsilon), float(self.cfg.policy.l2coeff),
                1.0 / rankers[i].n_fits_ranked, self._stream()), "es_adam_step")
            st.sync_host()

            ob = ObStat((self.env.ob_dims[i],), 0)
            ob.inc(ob_sums[i].cpu().numpy(), ob_sumsqs[i].cpu().numpy(),
                   float(ob_count.item()))
            ob.dist_inc(self.comm)
            obstats.append(ob)
            noiseless.append(float(fits[-1].item()))

        self.gen += 1
        self.timings = {"rollout_s": t1 - t0, "gen_s": time.perf_counter() - t0,
                        "env_steps": float(self.member_steps[:2 * self.pairs].sum().item())
                        * self.n_agents}
        return noiseless, obstats

    def update_obstats(self, obstats: List[ObStat]):
        for st, ob in zip(self.states, obstats):
            st.policy.update_obstat(ob)
            st.push_obstat()
