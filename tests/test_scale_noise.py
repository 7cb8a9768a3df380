"""Gradient reconstruction numerics (reference test/utils/utils_test.py:7-40):
batched == native == full dot, on an analytically predictable arange table."""
import numpy as np
import torch

from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.utils.utils import batch_noise, scale_noise


def _arange_table(n_params=10, size=200):
    return NoiseTable(n_params, torch.arange(size, dtype=torch.float32))


def test_batch_noise_slices():
    nt = _arange_table()
    inds = np.array([0, 5, 10, 15, 20, 25, 30])
    batches = list(batch_noise(inds, nt, 10, 3))
    assert [b.shape[0] for b in batches] == [3, 3, 1]  # ragged last batch
    flat = torch.cat(batches)
    for k, idx in enumerate(inds):
        assert torch.equal(flat[k], nt.noise[idx:idx + 10])


def test_scale_noise_matches_full_dot():
    nt = _arange_table()
    rs = np.random.RandomState(0)
    inds = rs.randint(0, 190, size=17)
    fits = rs.randn(17).astype(np.float32)
    got = scale_noise(fits, inds, nt, 10, batch_size=5).numpy()
    rows = np.stack([np.arange(i, i + 10, dtype=np.float32) for i in inds])
    expect = fits @ rows
    np.testing.assert_allclose(got, expect, rtol=1e-5)


def test_scale_noise_single_batch_equals_many():
    nt = _arange_table()
    inds = np.array([3, 50, 100])
    fits = np.array([1.0, -2.0, 0.5], dtype=np.float32)
    a = scale_noise(fits, inds, nt, 10, batch_size=1).numpy()
    b = scale_noise(fits, inds, nt, 10, batch_size=500).numpy()
    np.testing.assert_allclose(a, b, rtol=1e-6)
