#!/usr/bin/env python3
"""Minimal single-objective ES, written inline for didactic flexibility —
the equivalent of the reference's ``simple_example.py``.

Run (single process):       python examples/simple_example.py configs/simple_conf.json
Run (N CPU ranks, gloo):    torchrun --standalone --local-addr 127.0.0.1 \
                                --nproc-per-node N examples/simple_example.py \
                                configs/simple_conf.json

The inline generation block below is what ``es.step`` packages (reference
``simple_example.py:45-59``): test_params -> update obstat -> rank ->
approx_grad, with fitness triples exchanged over gloo/RCCL.

PROVENANCE: deliberately mirrors the reference's didactic script
(simple_example.py) line for line where the inline generation loop is the
point being demonstrated.
"""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import load_config, parse_args
from es_pytorch_amd.core import es
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import init_comm, seed_all
from es_pytorch_amd.rollout import RewardResult, run_model
from es_pytorch_amd.utils.rankers import CenteredRanker

if __name__ == "__main__":
    cfg = load_config(parse_args())
    comm = init_comm()

    env = make(cfg.env.name, max_steps=cfg.env.max_steps)
    # seeding before network creation so initial params replicate across ranks
    rs, my_seed, global_seed = seed_all(comm, cfg.general.seed)
    env.seed(my_seed)
    print(f"rank {comm.rank} seed:{my_seed} torch seed:{global_seed}")

    nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), env, cfg.policy.ac_std,
                     cfg.policy.ob_clip)
    policy = Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)), cfg.policy.lr))
    nt = NoiseTable.create_shared(comm, cfg.noise.tbl_size, len(policy),
                                  seed=cfg.general.seed)
    ranker = CenteredRanker()

    def r_fn(model: torch.nn.Module) -> RewardResult:
        save_obs = rs.random_sample() < cfg.policy.save_obs_chance
        rews, behv, obs, steps = run_model(model, env, cfg.env.max_steps, rs)
        return RewardResult(rews, behv,
                            obs if save_obs else np.array([np.zeros(env.observation_space.shape)]),
                            steps)

    assert cfg.general.policies_per_gen % comm.size == 0 and \
        (cfg.general.policies_per_gen / comm.size) % 2 == 0
    eps_per_proc = int((cfg.general.policies_per_gen / comm.size) / 2)

    for gen in range(cfg.general.gens):
        if comm.rank == 0:
            print(f"Generation:{gen}")

        gen_obstat = ObStat(env.observation_space.shape, 0)
        pos_fits, neg_fits, inds, steps = es.test_params(comm, eps_per_proc, policy, nt,
                                                         gen_obstat, r_fn, rs)
        policy.update_obstat(gen_obstat)
        ranker.rank(pos_fits, neg_fits, inds)
        es.approx_grad(policy, ranker, nt, policy.flat_params, cfg.general.batch_size,
                       cfg.policy.l2coeff)

        if comm.rank == 0:
            print(f"avg fitness:{np.mean(np.concatenate((pos_fits, neg_fits)))}\n")
        if gen % 10 == 0 and comm.rank == 0:
            policy.save(f"saved/{cfg.general.name}", str(gen))
