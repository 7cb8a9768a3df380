"""Episodic rollout runner (the CPU/eval path).

Mirrors the reference's single-episode loop (``src/gym/gym_runner.py:33-67``):
obs -> tensor -> model(ob, rs=rs) -> env.step -> collect rewards + 3-D body
position per step (the "behaviour" for novelty) + raw obs; behaviour padded
to 3*max_steps (``gym_runner.py:66``). Built-in envs expose ``position`` so
the four framework-specific position probes of the reference
(``gym_runner.py:13-30``) collapse to one accessor.

The GPU engine replaces this loop with population-batched on-device rollouts
(``core/engine.py``); this runner remains for CPU training, replay
(``examples/run_saved.py``) and tests.
"""
from __future__ import annotations

import time
from typing import Callable, List, Tuple

import numpy as np
import torch


def env_pos(env) -> Tuple[float, float, float]:
    """Position probe for built-in envs (replaces reference ``gym_runner.py:13-30``)."""
    p = getattr(env, "position", None)
    if p is None:
        return (0.0, 0.0, 0.0)
    p = np.asarray(p, dtype=np.float64).reshape(-1)
    return (float(p[0]), float(p[1]) if len(p) > 1 else 0.0, float(p[2]) if len(p) > 2 else 0.0)


def run_model(model: torch.nn.Module, env, max_steps: int,
              rs: np.random.RandomState = None, render: bool = False,
              get_pos_fn: Callable = env_pos) -> Tuple[List[float], List[float], np.ndarray, int]:
    """Single-episode rollout (reference ``gym_runner.py:33-67``).

    :returns: (rewards, behaviour, obs array, steps); behaviour is padded to
        length 3*max_steps by repeating the final position.
    """
    behv: List[float] = []
    rews: List[float] = []
    obs = []
    step = 0

    with torch.no_grad():
        ob = env.reset()
        for step in range(max_steps):
            ob_t = torch.from_numpy(np.asarray(ob)).float()
            action = model(ob_t, rs=rs)
            if isinstance(action, torch.Tensor):
                action = action.numpy()
            ob, rew, done, _ = env.step(action)
            rews.append(float(rew))
            obs.append(np.asarray(ob))
            behv.extend(get_pos_fn(getattr(env, "unwrapped", env)))

            if render:
                env.render()
                time.sleep(1 / 60)

            if done:
                break

    behv += behv[-3:] * (max_steps - int(len(behv) / 3))
    return rews, behv, np.array(obs), step


def multi_agent_runner(policies: List[torch.nn.Module], env, max_steps: int,
                       rs: np.random.RandomState = None, save_obs: bool = False,
                       render: bool = False):
    """Joint rollout of N policies in one shared env (reference ``gym_runner.py:70-111``).

    Each step every policy maps its own observation to an action; the env
    consumes the action list. Used by the co-evolution entry
    (``examples/multi_agent.py``).
    """
    rews, saved_obs, behv = [], [], []
    step = 0

    with torch.no_grad():
        obs = env.reset()
        for step in range(max_steps):
            actions = [policy(torch.from_numpy(np.asarray(ob)).float(), rs=rs)
                       for policy, ob in zip(policies, obs)]
            actions = [a.numpy() if isinstance(a, torch.Tensor) else a for a in actions]
            obs, rew, done, _ = env.step(actions)
            if save_obs:
                saved_obs.append(np.asarray(obs))
            rews.append(rew)
            behv.extend(env_pos(getattr(env, "unwrapped", env)))
            if render:
                env.render()
            if done:
                break

    if not saved_obs:
        saved_obs.append([np.zeros(np.shape(o)) for o in obs])

    behv += behv[-3:] * (max_steps - int(len(behv) / 3))
    return rews, behv, _stack_agent_obs(saved_obs), step


def _stack_agent_obs(saved_obs):
    """(steps, agents, ob_dim) array; heterogeneous teams (per-agent obs of
    different widths, e.g. a real Unity build with several behaviors) fall
    back to a (steps, agents) object array — MultiAgentTrainingResult's
    per-agent column reductions work on either."""
    try:
        return np.array(saved_obs)
    except ValueError:
        arr = np.empty((len(saved_obs), len(saved_obs[0])), dtype=object)
        for i, step_obs in enumerate(saved_obs):
            for j, o in enumerate(step_obs):
                arr[i, j] = np.asarray(o)
        return arr
