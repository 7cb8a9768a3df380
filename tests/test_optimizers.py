"""Optimizer numerics vs the reference formulas (reference src/nn/optimizers.py:28-61).

The reference never tested these (SURVEY.md §4 gap); asserted here against
independently hand-computed updates.
"""
import numpy as np
import pytest

from es_pytorch_amd.nn.optimizers import SGD, Adam, SimpleES


def test_simple_es():
    with pytest.warns(UserWarning, match="sign convention|descends"):
        o = SimpleES(3, lr=0.5)
    g = np.array([1.0, -2.0, 3.0], dtype=np.float32)
    np.testing.assert_allclose(o.step(g), 0.5 * g)
    assert o.t == 1


def test_sgd_momentum():
    o = SGD(2, lr=0.1, momentum=0.9)
    g1 = np.array([1.0, 2.0], dtype=np.float32)
    s1 = o.step(g1)
    # v1 = 0.1*g1 ; step = -lr*v1
    np.testing.assert_allclose(s1, -0.1 * (0.1 * g1), rtol=1e-6)
    g2 = np.array([-1.0, 0.5], dtype=np.float32)
    s2 = o.step(g2)
    v2 = 0.9 * (0.1 * g1) + 0.1 * g2
    np.testing.assert_allclose(s2, -0.1 * v2, rtol=1e-6)


def test_adam_bias_correction():
    lr, b1, b2, eps = 0.01, 0.9, 0.999, 1e-8
    o = Adam(2, lr=lr, beta1=b1, beta2=b2, epsilon=eps)
    g = np.array([0.5, -1.5], dtype=np.float32)
    m = v = np.zeros(2, dtype=np.float32)
    for t in range(1, 4):
        step = o.step(g)
        a = lr * np.sqrt(1 - b2 ** t) / (1 - b1 ** t)
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        expect = -a * m / (np.sqrt(v) + eps)
        np.testing.assert_allclose(step, expect, rtol=1e-5)


def test_state_roundtrip():
    o = Adam(4, lr=0.01)
    o.step(np.ones(4, dtype=np.float32))
    d = o.state_dict()
    o2 = Adam(4, lr=0.01)
    o2.load_state_dict(d)
    g = np.full(4, 0.3, dtype=np.float32)
    np.testing.assert_allclose(o.step(g.copy()), o2.step(g.copy()))
