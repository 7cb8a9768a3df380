"""Novelty math (reference test/utils/novelty_test.py semantics)."""
import numpy as np
import torch

from es_pytorch_amd.utils.novelty import novelty, novelty_batch, update_archive


def test_update_archive_serial():
    a = update_archive(None, [1.0, 2.0], None)
    np.testing.assert_array_equal(a, [[1.0, 2.0]])
    a = update_archive(None, [3.0, 4.0], a)
    np.testing.assert_array_equal(a, [[1.0, 2.0], [3.0, 4.0]])


def test_novelty_exact():
    archive = np.array([[0.0, 0.0], [3.0, 4.0], [6.0, 8.0]])
    b = np.array([0.0, 0.0])
    # dists: 0, 5, 10
    assert novelty(b, archive, 1) == 0.0
    assert novelty(b, archive, 2) == 2.5
    assert novelty(b, archive, 3) == 5.0
    # k greater than archive size -> all entries (reference test :27-33)
    assert novelty(b, archive, 10) == 5.0


def test_novelty_batch_matches_serial():
    rng = np.random.RandomState(0)
    archive = rng.randn(20, 2)
    behaviours = rng.randn(7, 2)
    nb = novelty_batch(torch.from_numpy(behaviours), torch.from_numpy(archive), 5)
    for i in range(7):
        assert abs(nb[i].item() - novelty(behaviours[i], archive, 5)) < 1e-5
