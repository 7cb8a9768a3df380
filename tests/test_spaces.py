"""Box/Discrete space semantics (gym-surface compatibility)."""
import numpy as np

from es_pytorch_amd.spaces import Box, Discrete


def test_box_shape_and_sample():
    b = Box(-1.0, 1.0, (3,))
    b.seed(0)
    s = b.sample()
    assert s.shape == (3,) and s.dtype == np.float32
    assert b.contains(s)
    assert not b.contains(np.array([2.0, 0.0, 0.0], dtype=np.float32))


def test_box_infinite_bounds_sample():
    b = Box(-np.inf, np.inf, (2,))
    b.seed(1)
    s = b.sample()  # samples from a bounded surrogate
    assert np.isfinite(s).all()


def test_box_broadcast_bounds():
    b = Box(np.array([-1.0, 0.0]), np.array([1.0, 2.0]))
    assert b.shape == (2,)
    assert b.low[1] == 0.0 and b.high[1] == 2.0


def test_discrete():
    d = Discrete(5)
    d.seed(2)
    for _ in range(10):
        assert d.contains(d.sample())
    assert not d.contains(5)
    assert not d.contains(-1)
