"""Polymorphic "what is fitness" result types.

The class names, constructor signature and property surface match the
reference hierarchy (``src/gym/training_result.py:9-97``) because the entry
scripts and ``es.step`` are generic over them; the bodies are this repo's own.

A rollout yields four raw artifacts — per-step rewards, a flat behaviour
trace of 3-D positions ``[x0,y0,z0, x1,y1,z1, ...]``, the observations kept
for normalization statistics, and the step count. Each subclass reduces those
to its fitness objective(s): total reward, per-step reward, planar distance,
x-displacement, novelty against an archive, or the 2-objective
[reward, novelty] pair that NSR/NSRA rank jointly.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import List, Tuple, Type

import numpy as np

from es_pytorch_amd.utils.novelty import novelty


class TrainingResult(ABC):
    """Raw artifacts of one policy evaluation + the fitness reduction."""

    def __init__(self, rewards: List[float], positions: List[float], obs: np.ndarray,
                 steps: int, *args, **kwargs):
        self.rewards = rewards
        self.positions = positions  # flat [x0,y0,z0, x1,y1,z1, ...]
        self.obs = obs
        self.steps = steps

    @property
    def ob_sum_sq_cnt(self) -> Tuple[np.ndarray, np.ndarray, int]:
        """(Σob, Σob², count) feeding the ObStat merge. An all-zero obs block
        means "nothing was saved this episode" (save_obs_chance miss) and
        must contribute count 0, not len(obs) zeros."""
        ob = np.asarray(self.obs)
        n = 0 if not ob.any() else ob.shape[0]
        return ob.sum(axis=0), (ob * ob).sum(axis=0), n

    @abstractmethod
    def get_result(self) -> List[float]:
        """The fitness objective vector this evaluation scored."""

    @property
    def result(self) -> List[float]:
        return self.get_result()

    @property
    def reward(self):
        return sum(self.rewards)

    @property
    def behaviour(self):
        """Final planar (x, y) — the novelty-search behaviour descriptor."""
        return self.positions[-3:-1]


class MultiAgentTrainingResult(TrainingResult):
    """Joint result of a co-evolution rollout: ``rewards`` and ``obs`` carry
    a trailing per-agent axis; reductions happen per column."""

    def get_result(self):
        return self.reward

    @property
    def reward(self):
        # one total-reward entry per agent
        return np.asarray(self.rewards).sum(axis=0).tolist()

    @property
    def ob_sum_sq_cnt(self):
        ob = np.asarray(self.obs)
        return [self._agent_stats(ob[:, a]) for a in range(ob.shape[1])]

    @staticmethod
    def _agent_stats(ob: np.ndarray) -> Tuple[np.ndarray, np.ndarray, int]:
        n = 0 if not ob.any() else ob.shape[0]
        return ob.sum(axis=0), (ob * ob).sum(axis=0), n

    def trainingresults(self, tr_type: Type[TrainingResult]) -> List[TrainingResult]:
        """Split into one single-agent result per agent (shared behaviour/steps)."""
        rews = np.asarray(self.rewards)
        obs = np.asarray(self.obs)
        return [tr_type(rews[:, a], self.positions, obs[:, a], self.steps)
                for a in range(rews.shape[1])]


class RewardResult(TrainingResult):
    """Fitness = total episode reward."""

    def get_result(self) -> List[float]:
        return [self.reward]


class MeanRewardResult(TrainingResult):
    """Fitness = reward per step survived (discourages early termination)."""

    def get_result(self) -> List[float]:
        return [self.reward / self.steps]


class DistResult(TrainingResult):
    """Fitness = planar distance of the final position from the origin."""

    def get_result(self) -> List[float]:
        x, y = self.positions[-3], self.positions[-2]
        return [float(np.hypot(x, y))]


class XDistResult(DistResult):
    """Fitness = signed x-displacement (directed locomotion)."""

    def get_result(self) -> List[float]:
        return [self.positions[-3]]


class NSResult(TrainingResult):
    """Fitness = novelty of the behaviour descriptor vs the archive."""

    def __init__(self, rewards, positions, obs, steps, archive: np.ndarray, k: int):
        super().__init__(rewards, positions, obs, steps)
        self.archive = archive
        self.k = k

    @property
    def novelty(self):
        return novelty(np.asarray(self.behaviour), self.archive, self.k)

    def get_result(self) -> List[float]:
        return [self.novelty]


class NSRResult(NSResult):
    """2-objective fitness [reward, novelty]; NSR/NSRA weight the two."""

    def get_result(self) -> List[float]:
        return [self.reward, self.novelty]
