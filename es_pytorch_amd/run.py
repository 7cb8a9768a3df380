"""Training-run assembly helpers shared by the entry scripts.

The reference's entry scripts each assemble comm + env + policy + noise table
by hand (``simple_example.py:17-44``, ``obj.py:21-50``, ``nsra.py:66-111``);
this module factors that assembly and adds the GPU/CPU path choice: on a CUDA
device the population-batched :class:`GpuEngine` runs whole generations
on-device; otherwise the episodic ``es.step`` path evaluates sequential
rollouts exactly like the reference.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.core import es
from es_pytorch_amd.core.engine import GpuEngine
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make, make_batched
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import Comm, init_comm, seed_all
from es_pytorch_amd.rollout import RewardResult, run_model
from es_pytorch_amd.utils.rankers import Ranker


def build_policy(cfg, env, comm: Comm) -> Policy:
    if "load" in cfg.policy:
        return Policy.load(cfg.policy.load)
    nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), env, cfg.policy.ac_std,
                     cfg.policy.ob_clip)
    return Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)), cfg.policy.lr))


def build_run(cfg: AttrDict, objective: str = "reward", use_gpu: Optional[bool] = None,
              env_kwargs: Optional[dict] = None):
    """Assemble (comm, rs, env, policy, nt, engine-or-None).

    engine is a GpuEngine when CUDA is available (or use_gpu forces it);
    otherwise the caller should use the episodic path (``episodic_step``).
    """
    comm = init_comm()
    use_gpu = torch.cuda.is_available() if use_gpu is None else use_gpu
    device = comm.device if use_gpu else torch.device("cpu")

    rs, my_seed, global_seed = seed_all(comm, cfg.general.seed)
    env_kwargs = dict(env_kwargs or {})
    env_kwargs.setdefault("max_steps", int(cfg.env.max_steps))

    if use_gpu:
        pairs = int(cfg.general.policies_per_gen // comm.size // 2)
        eps = max(1, int(cfg.general.get("eps_per_policy", 1) or 1))
        env = make_batched(cfg.env.name, (2 * pairs + 1) * eps, device, **env_kwargs)
    else:
        env = make(cfg.env.name, **env_kwargs)
        env.seed(my_seed)

    policy = build_policy(cfg, env, comm)
    nt = NoiseTable.create_shared(comm, int(cfg.noise.tbl_size), len(policy),
                                  seed=cfg.general.seed if cfg.general.seed is not None
                                  else global_seed, device=device if use_gpu else None)
    engine = None
    if use_gpu:
        engine = GpuEngine(cfg, comm, policy, nt, env, rs, objective=objective)
    return comm, rs, env, policy, nt, engine


def episodic_fit_fn(cfg, env, rs, result_type=RewardResult, archive_box: Optional[dict] = None):
    """Reference-style fit_fn closure (``obj.py:54-64``): one rollout, result
    object of the configured type; NSRResult pulls archive/k from archive_box."""

    def fit_fn(model: torch.nn.Module, use_ac_noise: bool = True):
        save_obs = rs.random_sample() < cfg.policy.save_obs_chance
        rews, behv, obs, steps = run_model(model, env, int(cfg.env.max_steps),
                                           rs if use_ac_noise else None)
        obs_out = obs if save_obs else np.array([np.zeros(env.observation_space.shape)])
        if archive_box is not None:
            return result_type(rews, behv, obs_out, steps, archive_box["archive"],
                               int(cfg.novelty.k))
        return result_type(rews, behv, obs_out, steps)

    return fit_fn


def step_any(cfg, comm, policy, nt, env, engine: Optional[GpuEngine], fit_fn,
             rs, ranker: Ranker, reporter) -> Tuple:
    """One generation via the engine (GPU) or the episodic core (CPU)."""
    if engine is not None:
        tr, gen_obstat = engine.step(ranker, reporter)
        engine.update_obstat(gen_obstat)
        return tr, gen_obstat
    tr, gen_obstat = es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, reporter)
    policy.update_obstat(gen_obstat)
    return tr, gen_obstat
