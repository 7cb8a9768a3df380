"""Minimal observation/action space types.

The reference sizes its networks from gym spaces
(``src/nn/nn.py:33,102-104``); gym is not a dependency of this framework, so
these two classes provide the same sizing/sampling surface for the built-in
environments.
"""
from __future__ import annotations

import numpy as np


class Box:
    def __init__(self, low, high, shape=None, dtype=np.float32):
        if shape is None:
            low_a = np.asarray(low)
            shape = low_a.shape if low_a.shape else (1,)
        shape = tuple(shape)
        self.low = np.broadcast_to(np.asarray(low, dtype=dtype), shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=dtype), shape).copy()
        self.shape = shape
        self.dtype = dtype
        self._rs = np.random.RandomState()

    def seed(self, seed=None):
        self._rs = np.random.RandomState(seed)

    def sample(self):
        lo = np.where(np.isfinite(self.low), self.low, -1.0)
        hi = np.where(np.isfinite(self.high), self.high, 1.0)
        return self._rs.uniform(lo, hi).astype(self.dtype)

    def contains(self, x):
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= self.low - 1e-6) and np.all(x <= self.high + 1e-6))

    def __repr__(self):
        return f"Box{self.shape}"


class Discrete:
    def __init__(self, n: int):
        self.n = int(n)
        self.shape = ()
        self.dtype = np.int64
        self._rs = np.random.RandomState()

    def seed(self, seed=None):
        self._rs = np.random.RandomState(seed)

    def sample(self):
        return int(self._rs.randint(self.n))

    def contains(self, x):
        return 0 <= int(x) < self.n

    def __repr__(self):
        return f"Discrete({self.n})"


class MultiDiscrete:
    def __init__(self, nvec):
        self.nvec = np.asarray(nvec, dtype=np.int64)
        self.shape = self.nvec.shape
        self.dtype = np.int64
        self._rs = np.random.RandomState()

    def seed(self, seed=None):
        self._rs = np.random.RandomState(seed)

    def sample(self):
        return (self._rs.random_sample(self.nvec.shape) * self.nvec).astype(np.int64)

    def contains(self, x):
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= 0) and np.all(x < self.nvec))

    def __repr__(self):
        return f"MultiDiscrete({self.nvec.tolist()})"


class Tuple:
    """A fixed tuple of component spaces (multi-agent obs/action surfaces)."""

    def __init__(self, spaces):
        self.spaces = tuple(spaces)
        self.shape = None

    def seed(self, seed=None):
        for i, s in enumerate(self.spaces):
            s.seed(None if seed is None else seed + i)

    def sample(self):
        return tuple(s.sample() for s in self.spaces)

    def contains(self, x):
        return len(x) == len(self.spaces) and all(s.contains(v)
                                                  for s, v in zip(self.spaces, x))

    def __len__(self):
        return len(self.spaces)

    def __getitem__(self, i):
        return self.spaces[i]

    def __repr__(self):
        return f"Tuple({list(self.spaces)})"
