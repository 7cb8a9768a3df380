"""Environment APIs.

The reference evaluates policies on gym/PyBullet/Unity environments, one
sequential episode per process (``src/gym/gym_runner.py``). This framework
has no gym dependency; it defines two surfaces:

* :class:`Env` — the episodic, gym-like CPU API (reset/step/seed/render)
  used by the reference-style runner and replay tools;
* :class:`BatchedEnv` — the MI355X-native API: B independent env instances
  stepped as one batch of torch tensors resident on device, capture-safe
  (fixed shapes, no host sync) so a whole rollout loop can be recorded into
  a hipGraph. Episodes that terminate early stay in the batch with their
  ``alive`` mask cleared (SURVEY.md §7.4 "batched heterogeneous rollouts").

``SingleFromBatched`` adapts any BatchedEnv (B=1, CPU) to the episodic API so
every environment is automatically available to both paths.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Optional, Tuple

import numpy as np
import torch


class Env(ABC):
    """Episodic gym-like environment (single instance, numpy, CPU)."""

    observation_space = None
    action_space = None

    @abstractmethod
    def reset(self) -> np.ndarray:
        ...

    @abstractmethod
    def step(self, action) -> Tuple[np.ndarray, float, bool, dict]:
        ...

    def seed(self, seed: Optional[int] = None):
        pass

    def render(self, mode: str = "human"):
        pass

    @property
    def position(self) -> Tuple[float, float, float]:
        """3-D body position — the behaviour probe (reference ``gym_runner.py:13-30``)."""
        return (0.0, 0.0, 0.0)

    @property
    def unwrapped(self):
        return self


class BatchedEnv(ABC):
    """B independent instances stepped as one device-resident batch."""

    def __init__(self, batch: int, device: torch.device):
        self.batch = int(batch)
        self.device = torch.device(device)

    observation_space = None
    action_space = None

    @property
    def ob_dim(self) -> int:
        return int(np.prod(self.observation_space.shape))

    @property
    def ac_dim(self) -> int:
        return int(np.prod(self.action_space.shape))

    @abstractmethod
    def reset(self, seed: Optional[int] = None) -> torch.Tensor:
        """:returns: obs (B, ob_dim) float32 on device."""

    @abstractmethod
    def step(self, actions: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Step all B instances; MUST be capture-safe (no host sync, fixed shapes).

        :returns: (obs (B, ob_dim), reward (B,), done (B,) bool) — reward is
            the raw per-step reward; alive-masking is the caller's job.
        """

    @property
    @abstractmethod
    def positions(self) -> torch.Tensor:
        """:returns: (B, 3) body positions (behaviour probe)."""


class SingleFromBatched(Env):
    """Adapt a BatchedEnv (B=1, CPU) to the episodic API."""

    def __init__(self, benv: BatchedEnv):
        assert benv.batch == 1
        self.benv = benv
        self.observation_space = benv.observation_space
        self.action_space = benv.action_space
        self._seed: Optional[int] = None

    def seed(self, seed: Optional[int] = None):
        self._seed = seed

    def reset(self) -> np.ndarray:
        ob = self.benv.reset(self._seed)
        if self._seed is not None:
            self._seed += 1  # new episode, new variation
        return ob[0].cpu().numpy()

    def step(self, action):
        a = torch.as_tensor(np.asarray(action, dtype=np.float32)).reshape(1, -1)
        ob, rew, done = self.benv.step(a)
        return ob[0].cpu().numpy(), float(rew[0]), bool(done[0]), {}

    @property
    def position(self):
        p = self.benv.positions[0].cpu().numpy()
        return (float(p[0]), float(p[1]), float(p[2]))
