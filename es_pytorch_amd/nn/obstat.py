"""Running observation statistics for virtual-batch-norm style normalization.

Same semantics as the reference (``src/nn/obstat.py:13-43``): running
(sum, sumsq, count) in fp64; ``std`` floored at 1e-2 (``obstat.py:37``); the
per-generation stat starts at eps=0 (``src/core/es.py:41``) and the lifetime
stat at eps=1e-2 (``src/core/policy.py:27``).

The cross-rank merge replaces the reference's custom pickling MPI reduce op
(``obstat.py:5-10,39-43``) with ONE packed fp64 all_reduce over
``[sum(ob_dim), sumsq(ob_dim), count]`` — RCCL-friendly, no pickling.


PROVENANCE: the running-stat math (1e-2 std floor, eps seeding semantics)
is the reference's (src/nn/obstat.py); the packed-tensor all_reduce merge
replacing the pickling MPI op is original.
"""
from __future__ import annotations

import numpy as np
import torch

from es_pytorch_amd.parallel.comm import Comm


class ObStat:
    def __init__(self, shape, eps: float):
        self.sum: np.ndarray = np.zeros(shape, dtype=np.float64)
        self.sumsq: np.ndarray = np.full(shape, eps, dtype=np.float64)
        self.count: float = eps

    def inc(self, s: np.ndarray, ssq: np.ndarray, c: float):
        self.sum += np.asarray(s, dtype=np.float64)
        self.sumsq += np.asarray(ssq, dtype=np.float64)
        self.count += c

    def __iadd__(self, other: "ObStat"):
        self.inc(other.sum, other.sumsq, other.count)
        return self

    def __repr__(self):
        return f"sum:{self.sum} sumsq:{self.sumsq} count:{self.count}"

    @property
    def mean(self) -> np.ndarray:
        return self.sum / self.count

    @property
    def std(self) -> np.ndarray:
        return np.sqrt(np.maximum(self.sumsq / self.count - np.square(self.mean), 1e-2))

    def dist_inc(self, comm: Comm):
        """Merge this stat across all ranks (reference ``mpi_inc``, ``obstat.py:39-43``).

        Packs [sum, sumsq, count] into one fp64 tensor and all-reduces it —
        the RCCL replacement for the pickling MPI op.
        """
        if comm is None or comm.size == 1:
            return
        flat = np.concatenate([self.sum.ravel(), self.sumsq.ravel(), [self.count]])
        t = torch.from_numpy(flat.copy())
        comm.allreduce_sum_(t)
        merged = t.numpy()
        n = self.sum.size
        self.sum = merged[:n].reshape(self.sum.shape)
        self.sumsq = merged[n:2 * n].reshape(self.sumsq.shape)
        self.count = float(merged[-1])

    # reference-compatible alias (``obstat.py:39``)
    mpi_inc = dist_inc
