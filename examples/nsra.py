#!/usr/bin/env python3
"""Novelty-search ES — the equivalent of the reference's ``nsra.py``:
NS-ES, NSR-ES, NSRA-ES and P-NSRA over a population of ``n_policies``
policies, one selected per generation proportional to novelty (or
round-robin for P-NSRA), 2-objective [reward, novelty] ranking via
MultiObjectiveRanker, an on-device behaviour archive, and the adaptive /
progressive reward-vs-novelty weight schedules (reference ``nsra.py:48-63``).

  python examples/nsra.py configs/nsra.json
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      examples/nsra.py configs/nsra.json
"""
import os
import random
import sys
from os import path
from typing import Tuple

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import load_config, parse_args
from es_pytorch_amd.core import es
from es_pytorch_amd.core.engine import GpuEngine
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make, make_batched
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import init_comm, seed_all
from es_pytorch_amd.rollout import NSRResult
from es_pytorch_amd.run import episodic_fit_fn
from es_pytorch_amd.utils.novelty import novelty, update_archive
from es_pytorch_amd.utils.rankers import CenteredRanker, MultiObjectiveRanker
from es_pytorch_amd.utils.reporters import (DefaultReporterSet, LoggerReporter,
                                            StdoutReporter)


def nsra_update(cfg, reward: float, obj_w: float, best_reward: float,
                time_since_best: int) -> Tuple[float, float, int]:
    """NSRA-ES weight schedule (reference ``nsra.py:48-63``)."""
    if reward > best_reward:
        return min(1.0, obj_w + cfg.nsr.weight_delta), reward, 0
    time_since_best += 1
    if time_since_best > cfg.nsr.max_time_since_best:
        obj_w = max(0.0, obj_w - cfg.nsr.weight_delta)
        time_since_best = 0
    return obj_w, best_reward, time_since_best


def main(cfg):
    comm = init_comm()
    use_gpu = torch.cuda.is_available()
    device = comm.device if use_gpu else torch.device("cpu")
    full_name = f"{cfg.env.name}-{cfg.general.name}"

    mlflow_reporter = None
    if cfg.general.get("mlflow"):
        from es_pytorch_amd.utils.reporters import MLFlowReporter
        mlflow_reporter = MLFlowReporter(comm, cfg)
    reporter = DefaultReporterSet(comm, full_name, LoggerReporter(comm, full_name),
                                  StdoutReporter(comm), mlflow_reporter)

    rs, my_seed, global_seed = seed_all(comm, cfg.general.seed)
    if cfg.nsr.adaptive:
        reporter.print("NSRA")
    elif cfg.nsr.progressive:
        reporter.print("P-NSRA")

    # population of policies (reference nsra.py:97-101)
    if use_gpu:
        pairs = int(cfg.general.policies_per_gen // comm.size // 2)
        eps = max(1, int(cfg.general.get("eps_per_policy", 1) or 1))
        env = make_batched(cfg.env.name, (2 * pairs + 1) * eps, device,
                           max_steps=cfg.env.max_steps)
    else:
        env = make(cfg.env.name, max_steps=cfg.env.max_steps)
        env.seed(my_seed)

    population, engines = [], []
    for _ in range(cfg.general.n_policies):
        nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), env, cfg.policy.ac_std,
                         cfg.policy.ob_clip)
        population.append(Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)),
                                                         cfg.policy.lr)))
    nt = NoiseTable.create_shared(comm, cfg.noise.tbl_size, len(population[0]),
                                  reporter, cfg.general.seed,
                                  device=device if use_gpu else None)
    if use_gpu:
        engines = [GpuEngine(cfg, comm, p, nt, env, rs, objective="nsr",
                             novelty_k=cfg.novelty.k) for p in population]

    archive = None
    archive_box = {"archive": None}

    def ns_fn(model, use_ac_noise=True):
        return episodic_fit_fn(cfg, env, rs, NSRResult, archive_box)(model, use_ac_noise)

    # atomic ring checkpointing + exact resume (utils/checkpoint.py): the
    # whole NSRA control state (archive, per-policy weights/novelties,
    # selection RNG) survives a kill (cfg.general.ckpt_every)
    ck_every = int(cfg.general.get("ckpt_every", 0) or 0)
    ckpt, start_gen, ck_state = None, 0, None
    if ck_every:
        from es_pytorch_amd.utils.checkpoint import RunCheckpointer
        ckpt = RunCheckpointer(path.join("saved", full_name, "ckpt"), comm,
                               keep=int(cfg.general.get("ckpt_keep", 3)),
                               every=ck_every)
        ck_state = ckpt.load()

    # ---- archive init (reference nsra.py:31-45) ----------------------------
    policies_novelties = []
    if ck_state is not None:
        pass  # resumed runs restore the archive instead of re-initializing
    elif use_gpu:
        # evaluation-only noiseless episode per policy (reference
        # init_archive, nsra.py:31-45: no training step, no RNG consumption)
        for eng in engines:
            _, b3, _ = eng.noiseless_eval()
            b = comm.broadcast_obj([float(b3[0]), float(b3[1])], src=0)
            archive = update_archive(None, b, archive)
            policies_novelties.append(max(1e-2, novelty(np.array(b), archive,
                                                        cfg.novelty.k)))
    else:
        for p in population:
            b = None
            if comm.rank == 0:
                behvs = [ns_fn(p.pheno(np.zeros(len(p))), False).behaviour
                         for _ in range(cfg.novelty.rollouts)]
                b = np.mean(behvs, axis=0)
            archive = update_archive(comm, b, archive)
            b = archive[-1]
            policies_novelties.append(max(1e-2, novelty(b, archive, cfg.novelty.k)))
    archive_box["archive"] = archive

    policies_best_rewards = [-np.inf] * cfg.general.n_policies
    time_since_best = [0] * cfg.general.n_policies
    obj_weight = [cfg.nsr.initial_w] * cfg.general.n_policies
    best_rew, best_dist = -np.inf, -np.inf

    if ck_state is not None:
        start_gen, extra = ckpt.restore(ck_state, population, rs, cfg=cfg,
                                        engine=engines or None)
        archive = np.asarray(extra["archive"])
        archive_box["archive"] = archive
        policies_novelties = extra["policies_novelties"]
        policies_best_rewards = extra["policies_best_rewards"]
        time_since_best = extra["time_since_best"]
        obj_weight = extra["obj_weight"]
        best_rew, best_dist = extra["best_rew"], extra["best_dist"]
        reporter.print(f"resumed from checkpoint at gen {start_gen}")

    for gen in range(start_gen, cfg.general.gens):
        idx = random.choices(range(len(policies_novelties)),
                             weights=policies_novelties, k=1)[0]
        if cfg.nsr.progressive:
            idx = gen % cfg.general.n_policies
        idx = comm.broadcast_obj(idx, src=0)
        ranker = MultiObjectiveRanker(CenteredRanker(), obj_weight[idx])
        if mlflow_reporter is not None:
            mlflow_reporter.set_active_run(idx)
        reporter.start_gen()
        reporter.log({"idx": idx})
        reporter.log({"w": obj_weight[idx]})
        reporter.log({"time since best": time_since_best[idx]})

        if use_gpu:
            eng = engines[idx]
            eng.archive = torch.from_numpy(np.asarray(archive)).to(device)
            tr, gen_obstat = eng.step(ranker, reporter)
            # archive growth + novelty re-score on device (ONE 2-float
            # broadcast); host mirror kept for checkpointing/saving
            nov = eng.grow_archive()
            archive = eng.archive.cpu().numpy()
            behv = list(archive[-1])
        else:
            tr, gen_obstat = es.step(cfg, comm, population[idx], nt, env, ns_fn, rs,
                                     ranker, reporter)
            tr = comm.broadcast_obj(tr, src=0)
            behv = comm.broadcast_obj(
                np.mean([ns_fn(population[idx].pheno(np.zeros(len(population[idx]))),
                               False).behaviour
                         for _ in range(cfg.novelty.rollouts)], axis=0), src=0)
        for i, policy in enumerate(population):  # shared obstat (nsra.py:127-128)
            policy.update_obstat(gen_obstat)
            if use_gpu:
                engines[i]._push_obstat()

        if not use_gpu:
            nov = comm.broadcast_obj(novelty(np.asarray(behv), archive, cfg.novelty.k),
                                     src=0)
            archive = update_archive(comm, behv, archive)
        archive_box["archive"] = archive
        policies_novelties[idx] = nov

        dist = float(np.linalg.norm(np.array(tr.positions[-3:-1])))
        rew = float(tr.reward) if not isinstance(tr.reward, list) else float(tr.reward[0])

        if cfg.nsr.adaptive:
            obj_weight[idx], policies_best_rewards[idx], time_since_best[idx] = \
                nsra_update(cfg, rew, obj_weight[idx], policies_best_rewards[idx],
                            time_since_best[idx])
        elif cfg.nsr.progressive:
            obj_weight[idx] = 1 if gen > cfg.nsr.end_progression_gen else \
                gen / cfg.nsr.end_progression_gen

        if (rew > best_rew or dist > best_dist) and comm.rank == 0:
            best_rew, best_dist = max(rew, best_rew), max(dist, best_dist)
            archive_path = path.join("saved", full_name, "archives")
            os.makedirs(archive_path, exist_ok=True)
            np.save(path.join(archive_path, f"{gen}.np"), archive)

        reporter.end_gen()

        if ckpt is not None:
            ckpt.maybe_save(gen + 1, population, rs, cfg=cfg,
                            engine=engines or None,
                            extra={"archive": np.asarray(archive),
                                   "policies_novelties": policies_novelties,
                                   "policies_best_rewards": policies_best_rewards,
                                   "time_since_best": time_since_best,
                                   "obj_weight": obj_weight,
                                   "best_rew": best_rew, "best_dist": best_dist})


if __name__ == "__main__":
    main(load_config(parse_args()))
