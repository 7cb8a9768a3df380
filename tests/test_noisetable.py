"""Noise table: determinism, sampling bounds, slice views
(reference test/es/noisetable_test.py semantics, CPU Philox twin)."""
import numpy as np
import pytest
import torch

from es_pytorch_amd.core.noisetable import NoiseTable


def test_make_noise_deterministic():
    a = NoiseTable.make_noise(100_000, seed=42)
    b = NoiseTable.make_noise(100_000, seed=42)
    assert torch.equal(a, b)
    c = NoiseTable.make_noise(100_000, seed=43)
    assert not torch.equal(a, c)


def test_make_noise_is_standard_normal():
    a = NoiseTable.make_noise(1_000_000, seed=7).numpy()
    assert abs(a.mean()) < 5e-3
    assert abs(a.std() - 1.0) < 5e-3
    # tail sanity: Box-Muller should produce |z|>4 at roughly the right rate
    assert 0 < (np.abs(a) > 4).sum() < 200


def test_prefix_stability():
    """Element i depends only on (seed, i), not on table size."""
    a = NoiseTable.make_noise(1000, seed=5)
    b = NoiseTable.make_noise(64, seed=5)
    assert torch.equal(a[:64], b)


def test_sample_and_get():
    nt = NoiseTable(10, NoiseTable.make_noise(1000, seed=1))
    rs = np.random.RandomState(3)
    idx, noise = nt.sample(rs)
    assert noise.shape == (10,)
    assert 0 <= idx < 990
    assert torch.equal(noise, nt.noise[idx:idx + 10])
    assert torch.equal(nt[idx], noise)
    with pytest.raises(AssertionError):
        nt.get(995, 10)


def test_sample_idxs_batched():
    nt = NoiseTable(10, NoiseTable.make_noise(1000, seed=1))
    rs = np.random.RandomState(3)
    idxs = nt.sample_idxs(rs, 100)
    assert idxs.shape == (100,)
    assert idxs.min() >= 0 and idxs.max() < 990
    # same rs state -> same draws as sequential sample_idx
    rs2 = np.random.RandomState(3)
    seq = np.array([nt.sample_idx(rs2, 10) for _ in range(100)])
    np.testing.assert_array_equal(idxs, seq)


def test_too_large_network():
    nt = NoiseTable(2000, NoiseTable.make_noise(1000, seed=1))
    with pytest.raises(ValueError):
        nt.sample(np.random.RandomState(0))
