#!/usr/bin/env python3
"""Goal-directed locomotion ("flagrun") — the equivalent of the reference's
``flagrun.py``: a goal-conditioned policy steered toward randomly re-sampled
flag positions. The reference builds a custom goal-concatenating net
(``flagrun.py:39-59``) and a multi-episode-averaging rollout
(``flagrun.py:80-142``); here the goal-conditioned observation is part of the
HumanoidFlagrun environment itself (``envs/locomotion.py``: obs =
[state, (goal-pos)*0.1]) so the standard MLP + engine path applies, with the
same goal-progress reward semantics.

  python examples/flagrun.py configs/flagrun.json
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import load_config, parse_args
from es_pytorch_amd.run import build_run, episodic_fit_fn, step_any
from es_pytorch_amd.utils.rankers import CenteredRanker
from es_pytorch_amd.utils.reporters import (DefaultReporterSet, LoggerReporter,
                                            StdoutReporter)


def main(cfg):
    comm, rs, env, policy, nt, engine = build_run(
        cfg, objective="reward", env_kwargs={"goal_conditioned": True}
        if "Flagrun" in cfg.env.name else None)

    full_name = f"{cfg.env.name}-{cfg.general.name}"
    reporter = DefaultReporterSet(comm, full_name, LoggerReporter(comm, full_name),
                                  StdoutReporter(comm))
    ranker = CenteredRanker()
    fit_fn = episodic_fit_fn(cfg, env, rs) if engine is None else None

    for gen in range(cfg.general.gens):
        reporter.start_gen()
        tr, _ = step_any(cfg, comm, policy, nt, env, engine, fit_fn, rs, ranker, reporter)
        cfg.noise.std = policy.std = max(cfg.noise.std * cfg.noise.get("std_decay", 1),
                                         cfg.noise.get("std_limit", 0))
        cfg.policy.lr = policy.optim.lr = max(cfg.policy.lr * cfg.policy.get("lr_decay", 1),
                                              cfg.policy.get("lr_limit", 0))
        reporter.end_gen()
        if gen % 10 == 0 and comm.rank == 0:
            if engine is not None:
                engine.sync_host()
            policy.save(f"saved/{full_name}", str(gen))

    if engine is not None:
        engine.sync_host()


if __name__ == "__main__":
    main(load_config(parse_args()))
