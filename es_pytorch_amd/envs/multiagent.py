"""Multi-agent co-evolution environment.

The reference wraps Unity ML-Agents sims for co-evolution
(``src/gym/unity.py:14-118``: per-team observation/action tuples, one env
process per rank). Unity is not available offline, so this module provides a
self-contained multi-agent game with the same interface shape the
co-evolution loop needs (reference ``multi_agent.py:33-67``): ``reset() ->
[obs_per_agent]``, ``step([action_per_agent]) -> ([obs], [rew], done, info)``.

The built-in game is a pursuit-evasion tag on a bounded 2-D arena: agent 0
(chaser) is rewarded for closing distance to agent 1 (runner), which is
rewarded for keeping away — strictly competitive, so co-evolution has
pressure in both directions.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np

from es_pytorch_amd.spaces import Box


class PursuitTag:
    """2-agent pursuit-evasion tag (episodic, numpy)."""

    N_AGENTS = 2
    ARENA = 5.0

    def __init__(self, max_steps: int = 200):
        self.max_steps = max_steps
        # obs per agent: own pos(2), own vel(2), other pos(2), other vel(2)
        self.observation_space = [Box(-np.inf, np.inf, (8,)) for _ in range(2)]
        self.action_space = [Box(-1.0, 1.0, (2,)) for _ in range(2)]
        self._rs = np.random.RandomState()
        self.p = np.zeros((2, 2))
        self.v = np.zeros((2, 2))
        self.t = 0

    def seed(self, seed: Optional[int] = None):
        self._rs = np.random.RandomState(seed)

    def _obs(self) -> List[np.ndarray]:
        return [np.concatenate([self.p[i], self.v[i], self.p[1 - i], self.v[1 - i]])
                .astype(np.float32) for i in range(2)]

    def reset(self) -> List[np.ndarray]:
        self.p = self._rs.uniform(-self.ARENA / 2, self.ARENA / 2, size=(2, 2))
        self.v = np.zeros((2, 2))
        self.t = 0
        return self._obs()

    def step(self, actions: List[np.ndarray]):
        dt = 0.1
        for i in range(2):
            a = np.clip(np.asarray(actions[i]).reshape(-1)[:2], -1, 1)
            self.v[i] = 0.8 * self.v[i] + a * dt * 5.0
        self.p = np.clip(self.p + self.v * dt, -self.ARENA, self.ARENA)
        self.t += 1

        d = float(np.linalg.norm(self.p[0] - self.p[1]))
        caught = d < 0.3
        # competitive shaping: chaser wants d small, runner wants d large
        rews = np.array([-d + (10.0 if caught else 0.0),
                         d - (10.0 if caught else 0.0)], dtype=np.float64)
        done = caught or self.t >= self.max_steps
        return self._obs(), rews, done, {}

    @property
    def position(self):
        return (float(self.p[0, 0]), float(self.p[0, 1]), 0.0)

    @property
    def unwrapped(self):
        return self


def make_multiagent(name: str, **kwargs):
    if name in ("PursuitTag", "PursuitTag-v0", "Tag"):
        return PursuitTag(**kwargs)
    raise ValueError(f"unknown multi-agent env {name!r}")


class BatchedPursuitTag:
    """B independent PursuitTag games stepped as one device batch.

    Same dynamics as :class:`PursuitTag` (torch instead of numpy), shaped for
    the multi-agent GPU engine: ``step([actions_agent0, actions_agent1])``
    with (B, 2) tensors -> (obs list of (B, 8), rewards (B, 2), done (B,)).
    Capture-safe (fixed shapes, in-place state).
    """

    N_AGENTS = 2
    ARENA = 5.0

    def __init__(self, batch: int, device="cpu", max_steps: int = 200):
        import torch
        self.batch = int(batch)
        self.device = torch.device(device)
        self.max_steps = max_steps
        self.observation_space = [Box(-np.inf, np.inf, (8,)) for _ in range(2)]
        self.action_space = [Box(-1.0, 1.0, (2,)) for _ in range(2)]
        self.ob_dims = [8, 8]
        self.ac_dims = [2, 2]
        self.p = torch.zeros(batch, 2, 2, device=self.device)
        self.v = torch.zeros(batch, 2, 2, device=self.device)
        self._steps = torch.zeros(batch, device=self.device)

    def reset(self, seed=None):
        import torch
        g = torch.Generator(device=self.device)
        if seed is not None:
            g.manual_seed(int(seed))
        self.p.copy_((torch.rand(self.batch, 2, 2, generator=g, device=self.device)
                      - 0.5) * self.ARENA)
        self.v.zero_()
        self._steps.zero_()
        return self._obs()

    def _obs(self):
        import torch
        out = []
        for i in range(2):
            out.append(torch.cat([self.p[:, i], self.v[:, i],
                                  self.p[:, 1 - i], self.v[:, 1 - i]], dim=1))
        return out

    def step(self, actions):
        import torch
        dt = 0.1
        for i in range(2):
            a = actions[i].reshape(self.batch, -1)[:, :2].clamp(-1, 1)
            self.v[:, i] = 0.8 * self.v[:, i] + a * dt * 5.0
        self.p.copy_((self.p + self.v * dt).clamp(-self.ARENA, self.ARENA))
        self._steps += 1

        d = (self.p[:, 0] - self.p[:, 1]).norm(dim=1)
        caught = d < 0.3
        bonus = torch.where(caught, 10.0, 0.0)
        rews = torch.stack([-d + bonus, d - bonus], dim=1)
        done = caught | (self._steps >= self.max_steps)
        return self._obs(), rews, done

    @property
    def positions(self):
        import torch
        return torch.cat([self.p[:, 0],
                          torch.zeros(self.batch, 1, device=self.device)], dim=1)
