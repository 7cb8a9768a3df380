"""Fitness shaping (rank transforms).

Full ranker family of the reference (``src/utils/rankers.py:9-120``), same
template method: ``rank() = _pre_rank -> _rank -> _post_rank`` where the
post-rank computes the antithetic difference ``ranked[:n_pos] -
ranked[n_pos:]`` (``rankers.py:42-44``). CenteredRanker is the OpenAI-ES
standard argsort-rank scaled to [-0.5, 0.5] (``rankers.py:53-58``).

Shapes are tiny (population-sized), so this runs in numpy on every rank
redundantly — identical inputs give identical updates, preserving the
reference's no-parameter-broadcast design.


PROVENANCE: the rank transforms are numeric definitions taken from the
reference (src/utils/rankers.py) — OpenAI-ES centered rank, antithetic
post-rank difference, elite re-indexing, weighted multi-objective; the
math IS the spec, so the bodies match it closely by intent.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Optional

import numpy as np


def rank(x: np.ndarray) -> np.ndarray:
    """Ranks in [0, len(x)) (reference ``rankers.py:9-17``).

    STABLE sort so tied fitnesses (common early, when many members terminate
    with identical reward) rank identically to the engine's device fast path
    (torch.argsort(stable=True), ``core/engine.py``) — GPU-vs-host runs stay
    comparable. The reference uses numpy's default introsort, which permutes
    ties arbitrarily; rank VALUES are identical either way.
    """
    assert x.ndim == 1
    ranks = np.empty(len(x), dtype=int)
    ranks[x.argsort(kind="stable")] = np.arange(len(x))
    return ranks


class Ranker(ABC):
    """Ranks all fitnesses obtained in a generation (reference ``rankers.py:20-50``)."""

    def __init__(self):
        self.fits_pos: Optional[np.ndarray] = None
        self.fits_neg: Optional[np.ndarray] = None
        self.noise_inds: Optional[np.ndarray] = None
        self.ranked_fits: Optional[np.ndarray] = None
        self.n_fits_ranked: int = 0

    fits = property(lambda self: np.concatenate((self.fits_pos, self.fits_neg)))

    @abstractmethod
    def _rank(self, x: np.ndarray) -> np.ndarray:
        ...

    def _pre_rank(self, fits_pos: np.ndarray, fits_neg: np.ndarray, noise_inds: np.ndarray):
        self.fits_pos = fits_pos
        self.fits_neg = fits_neg
        self.noise_inds = noise_inds

    def _post_rank(self, ranked_fits: np.ndarray) -> np.ndarray:
        self.n_fits_ranked = ranked_fits.size
        return ranked_fits[:len(self.fits_pos)] - ranked_fits[len(self.fits_pos):]

    def rank(self, fits_pos: np.ndarray, fits_neg: np.ndarray, noise_inds: np.ndarray) -> np.ndarray:
        self._pre_rank(fits_pos, fits_neg, noise_inds)
        ranked = self._rank(self.fits)
        self.ranked_fits = self._post_rank(ranked)
        return self.ranked_fits


class CenteredRanker(Ranker):
    """argsort-rank scaled to [-0.5, 0.5] (reference ``rankers.py:53-58``)."""

    def _rank(self, x: np.ndarray) -> np.ndarray:
        y = rank(x.ravel()).reshape(x.shape).astype(np.float32)
        y /= (x.size - 1)
        y -= 0.5
        return np.squeeze(y)


class DoublePositiveCenteredRanker(CenteredRanker):
    """Positive centered ranks doubled (reference ``rankers.py:61-65``)."""

    def _rank(self, x: np.ndarray) -> np.ndarray:
        y = super()._rank(x)
        y[y > 0] *= 2
        return y


class MaxNormalizedRanker(Ranker):
    """Fits normalized to [-1, 1] by min/max (reference ``rankers.py:68-74``)."""

    def _rank(self, x: np.ndarray) -> np.ndarray:
        mn = np.min(x)
        y = x + (-mn if mn > 0 else mn)
        y = y / np.max(y)
        y = 2 * y - 1
        return np.squeeze(y)


class SemiCenteredRanker(Ranker):
    """Quadratic semi-centered transform (reference ``rankers.py:77-82``)."""

    def _rank(self, x: np.ndarray) -> np.ndarray:
        y = rank(x.ravel()).reshape(x.shape).astype(np.float32)
        s = x.size
        y = (((1 / s) * np.square(y + 0.29 * s)) / s) - 0.5
        return y


class EliteRanker(Ranker):
    """Keep the top elite_percent of ranked fits and re-index noise_inds
    (reference ``rankers.py:85-103``)."""

    def __init__(self, ranker: Ranker, elite_percent: float):
        super().__init__()
        assert 0 <= elite_percent <= 1
        self.ranker = ranker
        self.elite_percent = elite_percent

    def _rank(self, x: np.ndarray) -> np.ndarray:
        ranked = self.ranker._rank(self.fits)
        n_elite = max(1, int(ranked.size * self.elite_percent))
        elite_fit_inds = np.argpartition(ranked, -n_elite)[-n_elite:]
        self.noise_inds = self.noise_inds[elite_fit_inds % len(self.noise_inds)]
        return ranked[elite_fit_inds]

    def _post_rank(self, ranked_fits: np.ndarray) -> np.ndarray:
        # no antithetic subtraction for elites (reference ``rankers.py:100-103``)
        self.n_fits_ranked = ranked_fits.size
        return ranked_fits


class MultiObjectiveRanker(Ranker):
    """w * rank(obj0) + (1-w) * rank(obj1), per column (reference ``rankers.py:106-120``).

    Used by NSR-ES / NSRA-ES with objectives [reward, novelty].
    """

    def __init__(self, ranker: Ranker, w: float):
        assert 0.0 <= w <= 1.0
        super().__init__()
        self.ranker = ranker
        self.w = w

    def _rank(self, x: np.ndarray) -> np.ndarray:
        assert x.shape[1] == 2, "MultiObjectiveRanker only supports 2 objectives"
        ranked = [self.ranker._rank(col) for col in x.T]
        return ranked[0] * self.w + ranked[1] * (1 - self.w)
