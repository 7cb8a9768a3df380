"""Flat-vector optimizers for the ES parameter update.

Same math as the reference (``src/nn/optimizers.py:7-61``, itself adapted from
uber-research/deep-neuroevolution): the optimizer operates on the FLAT
parameter vector and ``step(globalg)`` returns the delta to ADD to the params.

Sign convention (reference quirk, preserved deliberately): SGD and Adam return
the NEGATIVE step of their input (``optimizers.py:44,60``) and the caller
passes ``l2coeff*theta - grad`` (``src/core/es.py:101``), which nets out to
ascent on fitness with L2 decay. ``SimpleES`` returns ``+lr*g``
(``optimizers.py:33``) — sign-inconsistent with the other two; kept for API
parity but documented: use it only with a pre-negated input.

The CPU implementation below is numpy (matching reference numerics); the GPU
training engine uses the fused HIP Adam kernel (``ops/csrc/hip/update.hip``)
which implements the identical formula on device — parity-tested in
``tests/test_gpu_kernels.py``.


PROVENANCE: formulas ported verbatim from the reference (src/nn/optimizers
.py, itself adapted from uber-research/deep-neuroevolution) — the fused
HIP update kernel must match them bit-for-bit, so the host definitions
cannot drift.
"""
from __future__ import annotations

from abc import ABC, abstractmethod

import numpy as np


class Optimizer(ABC):
    def __init__(self, dim: int, lr: float):
        self.lr: float = lr
        self.dim: int = dim
        self.t: int = 0

    def step(self, globalg: np.ndarray) -> np.ndarray:
        """:returns: the delta to add to the flat params (reference ``optimizers.py:13-21``)."""
        self.t += 1
        return self._compute_step(globalg)

    @abstractmethod
    def _compute_step(self, globalg: np.ndarray) -> np.ndarray:
        ...

    def state_dict(self) -> dict:
        return {k: v for k, v in self.__dict__.items()}

    def load_state_dict(self, d: dict):
        self.__dict__.update(d)


class SimpleES(Optimizer):
    """Plain scaled-gradient step (reference ``optimizers.py:28-33``).

    WARNING (sign convention): returns ``+lr*g`` while SGD/Adam return the
    NEGATIVE step — swapping SimpleES into a caller that passes
    ``l2coeff*theta - grad`` (``es.approx_grad``) would DESCEND on fitness.
    A loud warning is emitted at construction; pass a pre-negated input or
    use SGD/Adam for the training path (GpuEngine rejects SimpleES)."""

    def __init__(self, dim: int, lr: float):
        super().__init__(dim, lr)
        import warnings
        warnings.warn(
            "SimpleES returns +lr*g (opposite sign convention to SGD/Adam); "
            "with the standard es.approx_grad input it descends on fitness — "
            "negate the input or use SGD/Adam for training",
            UserWarning, stacklevel=2)

    def _compute_step(self, globalg: np.ndarray) -> np.ndarray:
        return self.lr * globalg


class SGD(Optimizer):
    """Momentum SGD on the flat vector (reference ``optimizers.py:36-44``)."""

    def __init__(self, dim: int, lr: float, momentum: float = 0.9):
        super().__init__(dim, lr)
        self.v = np.zeros(self.dim, dtype=np.float32)
        self.momentum = momentum

    def _compute_step(self, globalg: np.ndarray) -> np.ndarray:
        self.v = self.momentum * self.v + (1.0 - self.momentum) * globalg
        return -self.lr * self.v


class Adam(Optimizer):
    """Adam on the flat vector (reference ``optimizers.py:47-61``)."""

    def __init__(self, dim: int, lr: float, beta1: float = 0.9, beta2: float = 0.999,
                 epsilon: float = 1e-08):
        super().__init__(dim, lr)
        self.beta1 = beta1
        self.beta2 = beta2
        self.epsilon = epsilon
        self.m = np.zeros(self.dim, dtype=np.float32)
        self.v = np.zeros(self.dim, dtype=np.float32)

    def _compute_step(self, globalgrad: np.ndarray) -> np.ndarray:
        a = self.lr * np.sqrt(1 - self.beta2 ** self.t) / (1 - self.beta1 ** self.t)
        self.m = self.beta1 * self.m + (1 - self.beta1) * globalgrad
        self.v = self.beta2 * self.v + (1 - self.beta2) * (globalgrad * globalgrad)
        return -a * self.m / (np.sqrt(self.v) + self.epsilon)
