"""ObStat semantics (reference src/nn/obstat.py + test/utils/obstat_test.py)."""
import numpy as np

from es_pytorch_amd.nn.obstat import ObStat


def test_inc_and_moments():
    s = ObStat((2,), eps=0)
    obs = np.array([[1.0, 2.0], [3.0, 6.0]])
    s.inc(obs.sum(0), (obs ** 2).sum(0), 2)
    np.testing.assert_allclose(s.mean, [2.0, 4.0])
    np.testing.assert_allclose(s.std, np.sqrt(np.maximum([1.0, 4.0], 1e-2)))


def test_std_floor():
    s = ObStat((1,), eps=0)
    s.inc(np.array([5.0]), np.array([25.0]), 1)  # zero variance
    np.testing.assert_allclose(s.std, [0.1])  # sqrt(1e-2) floor (reference obstat.py:37)


def test_iadd_merges():
    a = ObStat((2,), eps=1e-2)
    b = ObStat((2,), eps=0)
    b.inc(np.array([1.0, 1.0]), np.array([1.0, 1.0]), 3)
    a += b
    assert a.count == 1e-2 + 3
    np.testing.assert_allclose(a.sum, [1.0, 1.0])
