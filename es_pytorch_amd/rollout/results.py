"""Polymorphic "what is fitness" result types.

Same hierarchy as the reference (``src/gym/training_result.py:9-97``): a
rollout produces rewards, a behaviour trace (3-D positions per step), saved
observations and a step count; subclasses define the fitness objective(s) —
total reward, mean reward/step, distance, x-displacement, novelty, or the
2-objective [reward, novelty] used by NSR/NSRA.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import List, Tuple, Type

import numpy as np

from es_pytorch_amd.utils.novelty import novelty


class TrainingResult(ABC):
    """Result of a single policy evaluation (reference ``training_result.py:9-29``)."""

    def __init__(self, rewards: List[float], positions: List[float], obs: np.ndarray,
                 steps: int, *args, **kwargs):
        self.rewards: List[float] = rewards
        self.positions: List[float] = positions  # flat [x0,y0,z0, x1,y1,z1, ...]
        self.obs: np.ndarray = obs
        self.steps = steps

    @property
    def ob_sum_sq_cnt(self) -> Tuple[np.ndarray, np.ndarray, int]:
        cnt = len(self.obs) if np.any(self.obs) else 0
        return self.obs.sum(axis=0), np.square(self.obs).sum(axis=0), cnt

    @abstractmethod
    def get_result(self) -> List[float]:
        ...

    result: List[float] = property(lambda self: self.get_result())
    reward = property(lambda self: sum(self.rewards))
    behaviour = property(lambda self: self.positions[-3:-1])  # final (x, y)


class MultiAgentTrainingResult(TrainingResult):
    """Joint result of a co-evolution rollout (reference ``training_result.py:32-59``)."""

    def get_result(self):
        return self.reward

    @property
    def ob_sum_sq_cnt(self):
        out = []
        for i in range(self.obs.shape[1]):
            curr = self.obs[:, i]
            cnt = len(curr) if np.any(curr) else 0
            out.append((curr.sum(axis=0), np.square(curr).sum(axis=0), cnt))
        return out

    def trainingresults(self, tr_type: Type[TrainingResult]) -> List[TrainingResult]:
        rews, obs = np.array(self.rewards), np.array(self.obs)
        return [tr_type(rews[:, i], self.positions, obs[:, i], self.steps)
                for i in range(np.array(self.rewards).shape[1])]

    reward = property(lambda self: np.sum(self.rewards, axis=0).tolist())


class RewardResult(TrainingResult):
    def get_result(self) -> List[float]:
        return [self.reward]


class MeanRewardResult(TrainingResult):
    def get_result(self) -> List[float]:
        return [self.reward / self.steps]


class DistResult(TrainingResult):
    def get_result(self) -> List[float]:
        return [float(np.linalg.norm(self.positions[-3:-1]))]


class XDistResult(DistResult):
    def get_result(self) -> List[float]:
        return [self.positions[-3]]


class NSResult(TrainingResult):
    """Novelty-only fitness (reference ``training_result.py:82-92``)."""

    def __init__(self, rewards, positions, obs, steps, archive: np.ndarray, k: int):
        super().__init__(rewards, positions, obs, steps)
        self.archive = archive
        self.k = k

    novelty = property(lambda self: novelty(np.array(self.behaviour), self.archive, self.k))

    def get_result(self) -> List[float]:
        return [self.novelty]


class NSRResult(NSResult):
    """[reward, novelty] 2-objective fitness (reference ``training_result.py:95-97``)."""

    def get_result(self) -> List[float]:
        return [sum(self.rewards), self.novelty]
