#!/usr/bin/env python3
"""Multi-agent co-evolution — the equivalent of the reference's
``multi_agent.py``: N policies perturbed jointly, evaluated in one shared
env rollout, each updated from its own fitness column (reference
``multi_agent.py:33-67,110-131``). Uses the built-in PursuitTag env in place
of the reference's Unity sims (``envs/multiagent.py``).

  python examples/multi_agent.py configs/multi_agent.json
"""
import os
import sys
from typing import List

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import load_config, parse_args
from es_pytorch_amd.core import es
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs.multiagent import make_multiagent
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import init_comm, seed_all
from es_pytorch_amd.rollout import MultiAgentTrainingResult, RewardResult, multi_agent_runner
from es_pytorch_amd.utils.rankers import CenteredRanker
from es_pytorch_amd.utils.reporters import StdoutReporter


class _AgentEnvView:
    """Per-agent space view so FeedForward sizes from one agent's spaces."""

    def __init__(self, env, i):
        self.observation_space = env.observation_space[i]
        self.action_space = env.action_space[i]


def custom_test_params(comm, n: int, policies: List[Policy], nt: NoiseTable, env,
                       obstats: List[ObStat], rs, max_steps: int):
    """Joint evaluation: one noise draw per policy per eval, shared rollout
    (reference ``multi_agent.py:33-67``)."""
    n_agents = len(policies)
    results = [[] for _ in range(n_agents)]
    inds = [[] for _ in range(n_agents)]
    steps_total = 0
    for _ in range(n):
        noises = []
        for i, p in enumerate(policies):
            idx, noise = nt.sample(rs)
            inds[i].append(idx)
            noises.append(noise)
        models = [p.pheno(nz) for p, nz in zip(policies, noises)]
        rews, behv, obs, steps = multi_agent_runner(models, env, max_steps, rs,
                                                    save_obs=True)
        joint = MultiAgentTrainingResult(rews, behv, obs, steps)
        per_agent = joint.trainingresults(RewardResult)
        steps_total += steps
        for i in range(n_agents):
            results[i].append(per_agent[i])
            obstats[i].inc(*per_agent[i].ob_sum_sq_cnt)

    shared = []
    for i in range(n_agents):
        pos = [tr.result for tr in results[i]]
        neg = [tr.result for tr in results[i]]  # reference quirk preserved:
        # multi_agent.py:48-49 evaluates the SAME +noise nets twice; the
        # "negative" evaluation never subtracts noise
        rows = es._share_results(comm, pos, neg, inds[i])
        obstats[i].dist_inc(comm)
        shared.append(rows)
    steps_total = int(comm.allreduce_scalar(steps_total))
    return shared, steps_total


def main(cfg):
    comm = init_comm()
    rs, my_seed, _ = seed_all(comm, cfg.general.seed)
    if torch.cuda.is_available():
        return main_gpu(cfg, comm, rs)
    env = make_multiagent(cfg.env.name, max_steps=cfg.env.max_steps)
    env.seed(my_seed)
    reporter = StdoutReporter(comm)

    n_agents = env.N_AGENTS
    policies = []
    for i in range(n_agents):
        nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), _AgentEnvView(env, i),
                         cfg.policy.ac_std, cfg.policy.ob_clip)
        policies.append(Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)),
                                                       cfg.policy.lr)))
    nt = NoiseTable.create_shared(comm, cfg.noise.tbl_size, len(policies[0]),
                                  reporter, cfg.general.seed)
    obstats = [ObStat(env.observation_space[i].shape, 1e-2) for i in range(n_agents)]

    eps_per_proc = max(1, int(cfg.general.policies_per_gen / comm.size / 2))
    for gen in range(cfg.general.gens):
        reporter.start_gen()
        gen_obstats = [ObStat(env.observation_space[i].shape, 0) for i in range(n_agents)]
        shared, steps = custom_test_params(comm, eps_per_proc, policies, nt, env,
                                           gen_obstats, rs, cfg.env.max_steps)
        for i, policy in enumerate(policies):
            policy.update_obstat(gen_obstats[i])
            rows = shared[i]
            ranker = CenteredRanker()
            ranker.rank(rows[:, :1], rows[:, 1:2], rows[:, -1])
            es.approx_grad(policy, ranker, nt, policy.flat_params,
                           cfg.general.batch_size, cfg.policy.l2coeff)
            reporter.log({f"agent{i} avg": float(np.mean(rows[:, 0]))})
        reporter.log({"steps": steps})
        reporter.end_gen()
        if gen % 10 == 0 and comm.rank == 0:
            for i, p in enumerate(policies):
                p.save(f"saved/{cfg.general.name}/agent{i}", str(gen))


def main_gpu(cfg, comm, rs):
    """GPU-batched co-evolution (beyond reference parity): all perturbations
    of every policy play each other in batched env instances."""
    from es_pytorch_amd.core.ma_engine import MultiAgentGpuEngine
    from es_pytorch_amd.envs.multiagent import BatchedPursuitTag

    reporter = StdoutReporter(comm)
    pairs = int(cfg.general.policies_per_gen // comm.size // 2)
    B = 2 * pairs + 1
    env = BatchedPursuitTag(B, comm.device, max_steps=cfg.env.max_steps)
    policies = []
    for i in range(env.N_AGENTS):
        nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), _AgentEnvView(env, i),
                         cfg.policy.ac_std, cfg.policy.ob_clip)
        policies.append(Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)),
                                                       cfg.policy.lr)))
    nt = NoiseTable.create_shared(comm, cfg.noise.tbl_size, len(policies[0]),
                                  reporter, cfg.general.seed, device=comm.device)
    engine = MultiAgentGpuEngine(cfg, comm, policies, nt, env, rs)
    for gen in range(cfg.general.gens):
        reporter.start_gen()
        rankers = [CenteredRanker() for _ in range(env.N_AGENTS)]
        noiseless, obstats = engine.step(rankers)
        engine.update_obstats(obstats)
        for i, r in enumerate(rankers):
            reporter.log({f"agent{i} avg": float(np.mean(r.fits[:, 0]))})
            reporter.log({f"agent{i} noiseless": noiseless[i]})
        reporter.log({"steps": engine.timings["env_steps"]})
        reporter.end_gen()
        if gen % 10 == 0 and comm.rank == 0:
            for i, p in enumerate(policies):
                p.save(f"saved/{cfg.general.name}/agent{i}", str(gen))


if __name__ == "__main__":
    main(load_config(parse_args()))
