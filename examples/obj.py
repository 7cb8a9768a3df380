#!/usr/bin/env python3
"""Production single-objective ES run — the equivalent of the reference's
``obj.py``: full reporter stack, checkpoint-resume, lr/sigma/action-std decay
schedules, stagnation-triggered noise increase and elite-percent switching,
best-perturbation saving.

On a CUDA machine the whole generation runs on-GPU via GpuEngine (batched
rollouts, HIP kernels, RCCL triples); on CPU it falls back to the episodic
reference-style path. Launch one process per GPU with torchrun.

  python examples/obj.py configs/obj.json
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      examples/obj.py configs/obj.json
"""
import os
import sys
from os import path

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import load_config, parse_args
from es_pytorch_amd.run import build_run, episodic_fit_fn, step_any
from es_pytorch_amd.utils.rankers import CenteredRanker, EliteRanker
from es_pytorch_amd.utils.reporters import (DefaultReporterSet, LoggerReporter,
                                            StdoutReporter)


def main(cfg):
    comm, rs, env, policy, nt, engine = build_run(cfg, objective="reward")

    full_name = f"{cfg.env.name}-{cfg.general.name}"
    mlflow_reporter = None
    if cfg.general.get("mlflow"):
        from es_pytorch_amd.utils.reporters import MLFlowReporter
        mlflow_reporter = MLFlowReporter(comm, cfg)
    reporter = DefaultReporterSet(comm, full_name, LoggerReporter(comm, full_name),
                                  StdoutReporter(comm), mlflow_reporter)

    nn = policy._module
    ranker = CenteredRanker()
    elite = float(cfg.experimental.get("elite", 0)) if "experimental" in cfg else 0
    if 0 < elite < 1:
        ranker = EliteRanker(CenteredRanker(), elite)

    fit_fn = episodic_fit_fn(cfg, env, rs) if engine is None else None

    best_max_rew = -np.inf
    time_since_best = 0
    noise_std_inc = 0.08

    # hang watchdog: a wedged collective or kernel turns into a loud exit
    # (code 124) restartable from the checkpoint ring, instead of silently
    # holding the node (utils/watchdog.py)
    wd = hb = None
    if cfg.general.get("gen_timeout_s"):
        from es_pytorch_amd.utils.watchdog import Heartbeat, Watchdog
        wd = Watchdog(float(cfg.general.gen_timeout_s))
        hb = Heartbeat(path.join("saved", full_name, "heartbeat"), comm.rank)

    # atomic ring checkpointing + exact resume (utils/checkpoint.py); a
    # relaunch of the same command continues bit-for-bit from the newest
    # snapshot (beyond the reference's policy-only checkpoints, obj.py:39-41)
    ck_every = int(cfg.general.get("ckpt_every", 0) or 0)
    ckpt, start_gen = None, 0
    if ck_every:
        from es_pytorch_amd.utils.checkpoint import RunCheckpointer
        ckpt = RunCheckpointer(path.join("saved", full_name, "ckpt"), comm,
                               keep=int(cfg.general.get("ckpt_keep", 3)),
                               every=ck_every)
        state = ckpt.load()
        if state is not None:
            start_gen, extra = ckpt.restore(
                state, policy, rs, cfg=cfg, engine=engine, env=env,
                allow_reshard=bool(cfg.general.get("ckpt_allow_reshard", False)))
            best_max_rew = extra.get("best_max_rew", best_max_rew)
            time_since_best = extra.get("time_since_best", time_since_best)
            if 0 < elite < 1:
                ranker.elite_percent = extra.get("elite_percent", ranker.elite_percent)
            reporter.print(f"resumed from checkpoint at gen {start_gen}")

    for gen in range(start_gen, cfg.general.gens):
        if mlflow_reporter is not None:
            mlflow_reporter.set_active_run(0)
        reporter.start_gen()

        # decay schedules (reference obj.py:71-83); cfg mutated like the reference
        if cfg.noise.get("std_decay", 1) != 1:
            reporter.log({"noise std": policy.std})
        if cfg.policy.get("lr_decay", 1) != 1:
            reporter.log({"lr": policy.optim.lr})
        if cfg.policy.get("ac_std_decay", 1) != 1:
            reporter.log({"ac std": nn._action_std})

        if wd is not None:
            with wd.guard(f"generation {gen}"):
                tr, gen_obstat = step_any(cfg, comm, policy, nt, env, engine,
                                          fit_fn, rs, ranker, reporter)
            hb.beat(gen)
        else:
            tr, gen_obstat = step_any(cfg, comm, policy, nt, env, engine, fit_fn,
                                      rs, ranker, reporter)
        if engine is not None:  # per-phase timers (rollout/collective/update)
            reporter.log({k: round(v, 4) for k, v in engine.timings.items()
                          if k.endswith("_s")})

        cfg.policy.ac_std = nn._action_std = nn._action_std * cfg.policy.get("ac_std_decay", 1)
        cfg.noise.std = policy.std = max(cfg.noise.std * cfg.noise.get("std_decay", 1),
                                         cfg.noise.get("std_limit", 0))
        cfg.policy.lr = policy.optim.lr = max(cfg.policy.lr * cfg.policy.get("lr_decay", 1),
                                              cfg.policy.get("lr_limit", 0))

        reporter.log({"obs recorded": policy.obstat.count})

        fits = np.atleast_2d(ranker.fits)
        max_rew_ind = int(np.argmax(fits[:, 0]))
        max_rew = float(fits[:, 0][max_rew_ind])
        time_since_best = 0 if max_rew > best_max_rew else time_since_best + 1
        reporter.log({"time since best": time_since_best})

        exp = cfg.get("experimental", {})
        if time_since_best > exp.get("max_time_since_best", np.inf) and \
                exp.get("explore_with_large_noise", False):
            cfg.noise.std = policy.std = policy.std + noise_std_inc

        if 0 < elite < 1:  # elite extension (reference obj.py:96-101)
            if time_since_best > exp.get("max_time_since_best", np.inf):
                ranker.elite_percent = elite
            if time_since_best == 0:
                ranker.elite_percent = 1
            reporter.print(f"elite percent: {ranker.elite_percent}")

        # save best single perturbed individual (reference obj.py:104-110)
        if max_rew > best_max_rew and comm.rank == 0:
            best_max_rew = max_rew
            if engine is not None:
                engine.sync_host()
            coeff = 1 if max_rew_ind < ranker.n_fits_ranked // 2 else -1
            idx = int(ranker.noise_inds[max_rew_ind % max(1, ranker.n_fits_ranked // 2)])
            # engine noise indexes forward-layout params; permute to flat
            noise = engine.noise_slice_flat(idx) if engine is not None \
                else nt.get(idx, len(policy))
            folder = path.join("saved", full_name, "weights")
            os.makedirs(folder, exist_ok=True)
            torch.save(policy.pheno(coeff * noise),
                       path.join(folder, f"gen{gen}-rew{best_max_rew:0.0f}.pt"))
            policy.pheno()  # restore module to unperturbed params
            reporter.print(f"saving max policy with rew:{best_max_rew:0.2f}")

        reporter.end_gen()

        if ckpt is not None:
            extra = {"best_max_rew": best_max_rew, "time_since_best": time_since_best}
            if 0 < elite < 1:
                extra["elite_percent"] = ranker.elite_percent
            ckpt.maybe_save(gen + 1, policy, rs, cfg=cfg, engine=engine, env=env,
                            extra=extra)

    if engine is not None:
        engine.sync_host()


if __name__ == "__main__":
    main(load_config(parse_args()))
