"""run.py assembly helpers on the CPU path."""
import numpy as np
import torch

from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.rollout import NSRResult, RewardResult
from es_pytorch_amd.run import build_run, episodic_fit_fn, step_any
from es_pytorch_amd.utils.rankers import CenteredRanker
from es_pytorch_amd.utils.reporters import StdoutReporter


def _cfg():
    return AttrDict({
        "env": {"name": "CartPole-v1", "max_steps": 30},
        "noise": {"tbl_size": 100_000, "std": 0.05},
        "policy": {"layer_sizes": [8], "ac_std": 0.01, "l2coeff": 0.005, "lr": 0.02,
                   "ob_clip": 5, "save_obs_chance": 1.0},
        "general": {"name": "t", "gens": 1, "policies_per_gen": 4, "batch_size": 100,
                    "seed": 5},
        "novelty": {"k": 3},
    })


def test_build_run_cpu():
    cfg = _cfg()
    comm, rs, env, policy, nt, engine = build_run(cfg, use_gpu=False)
    assert engine is None  # CPU -> episodic path
    assert env.observation_space.shape == (4,)
    assert len(nt.noise) == 100_000
    assert len(policy) == 4 * 8 + 8 + 8 * 1 + 1


def test_episodic_fit_fn_and_step_any():
    cfg = _cfg()
    comm, rs, env, policy, nt, engine = build_run(cfg, use_gpu=False)
    fit_fn = episodic_fit_fn(cfg, env, rs)
    tr = fit_fn(policy.pheno())
    assert isinstance(tr, RewardResult)
    assert tr.steps > 0
    flat0 = policy.flat_params.copy()
    ranker = CenteredRanker()
    tr2, obstat = step_any(cfg, comm, policy, nt, env, None, fit_fn, rs, ranker,
                           StdoutReporter(comm))
    assert not np.array_equal(policy.flat_params, flat0)
    assert policy.obstat.count > 1e-2  # step_any folded the gen obstat in


def test_episodic_fit_fn_nsr():
    cfg = _cfg()
    comm, rs, env, policy, nt, engine = build_run(cfg, use_gpu=False)
    box = {"archive": np.array([[0.0, 0.0], [1.0, 1.0]])}
    fit_fn = episodic_fit_fn(cfg, env, rs, NSRResult, box)
    tr = fit_fn(policy.pheno())
    r = tr.result
    assert len(r) == 2  # [reward, novelty]
    assert r[1] >= 0
