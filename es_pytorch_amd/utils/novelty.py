"""Novelty-search support: behaviour archive + k-nearest-neighbour novelty.

Same math as the reference (``src/utils/novelty.py:9-18``): the archive is a
growing (N, d) array of behaviours; novelty = mean Euclidean distance to the
k nearest archive entries; new entries are produced on rank 0 and broadcast
(reference ``comm.scatter([b]*size)`` -> RCCL broadcast).

``novelty_batch`` is the population-engine version: novelty of B behaviours
against the archive in one torch cdist + topk (on device when the archive
tensor lives there) — SURVEY.md K8.


PROVENANCE: novelty/update_archive reproduce the reference math
(src/utils/novelty.py); the batched on-device variants are original.
"""
from __future__ import annotations

from typing import Optional, Sequence

import numpy as np
import torch

from es_pytorch_amd.parallel.comm import Comm


def update_archive(comm: Optional[Comm], behaviour: Sequence[float],
                   archive: Optional[np.ndarray]) -> np.ndarray:
    """Broadcast rank-0's new behaviour and append it (reference ``novelty.py:9-13``)."""
    if comm is not None:
        behaviour = comm.broadcast_obj(behaviour, src=0)
    behaviour = np.asarray(behaviour, dtype=np.float64)
    if archive is None:
        return np.array([behaviour])
    return np.concatenate((archive, [behaviour]))


def novelty(behaviour: np.ndarray, archive: np.ndarray, n: int) -> float:
    """Mean distance to the n nearest archive entries (reference ``novelty.py:16-18``)."""
    d = np.linalg.norm(np.asarray(archive, dtype=np.float64) -
                       np.asarray(behaviour, dtype=np.float64)[None, :], axis=1)
    k = min(n, len(d))
    return float(np.mean(np.partition(d, k - 1)[:k]))


def novelty_batch(behaviours: torch.Tensor, archive: torch.Tensor, n: int) -> torch.Tensor:
    """Novelty of (B, d) behaviours vs an (N, d) archive; stays on device.

    :returns: (B,) float32 novelty scores
    """
    d = torch.cdist(behaviours.to(archive.dtype), archive)  # (B, N)
    k = min(n, archive.shape[0])
    smallest, _ = torch.topk(d, k, dim=1, largest=False)
    return smallest.mean(dim=1).float()
