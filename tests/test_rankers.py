"""Ranker family vs hand-computed values (reference src/utils/rankers.py).

Includes the reference's MultiObjective test (reference test/utils/rankers.py:6-27
— which was never collected there due to its filename; fixed here) plus the
coverage the reference lacked.
"""
import numpy as np

from es_pytorch_amd.utils.rankers import (CenteredRanker, DoublePositiveCenteredRanker,
                                          EliteRanker, MaxNormalizedRanker,
                                          MultiObjectiveRanker, rank)


def test_rank_basic():
    x = np.array([3.0, 1.0, 2.0])
    np.testing.assert_array_equal(rank(x), [2, 0, 1])


def test_centered_ranker_values():
    r = CenteredRanker()
    pos = np.array([[1.0], [4.0]])
    neg = np.array([[2.0], [3.0]])
    ranked = r.rank(pos, neg, np.array([10, 20]))
    # fits [1,4,2,3] -> ranks [0,3,1,2] -> /3 -.5 = [-0.5, 0.5, -1/6, 1/6]
    # post: pos - neg = [-0.5 - (-1/6), 0.5 - 1/6]
    np.testing.assert_allclose(ranked, [-1 / 3, 1 / 3], atol=1e-6)
    assert r.n_fits_ranked == 4


def test_double_positive():
    r = DoublePositiveCenteredRanker()
    y = r._rank(np.array([1.0, 2.0, 3.0, 4.0, 5.0]))
    # centered: [-.5, -.25, 0, .25, .5] -> positives doubled
    np.testing.assert_allclose(y, [-0.5, -0.25, 0.0, 0.5, 1.0], atol=1e-6)


def test_max_normalized():
    r = MaxNormalizedRanker()
    y = r._rank(np.array([2.0, 4.0, 6.0]))
    # min>0 -> shift to 0: [0,2,4] -> /4 -> [0,.5,1] -> *2-1 = [-1,0,1]
    np.testing.assert_allclose(y, [-1.0, 0.0, 1.0])


def test_moo_weighted_rank():
    """Reference test/utils/rankers.py:6-27 semantics."""
    pos = np.array([[1.0, 10.0], [3.0, 30.0]])
    neg = np.array([[2.0, 20.0], [4.0, 40.0]])
    inds = np.array([1, 2])
    for w in (0.5, 0.1):
        moo = MultiObjectiveRanker(CenteredRanker(), w)
        got = moo.rank(pos, neg, inds)
        c = CenteredRanker()
        fits = np.concatenate((pos, neg))
        col0 = c._rank(fits[:, 0])
        col1 = c._rank(fits[:, 1])
        combined = col0 * w + col1 * (1 - w)
        expect = combined[:2] - combined[2:]
        np.testing.assert_allclose(got, expect, atol=1e-6)


def test_elite_ranker_no_antithetic_diff():
    e = EliteRanker(CenteredRanker(), 0.5)
    pos = np.array([[1.0], [5.0]])
    neg = np.array([[2.0], [6.0]])
    inds = np.array([100, 200])
    ranked = e.rank(pos, neg, inds)
    assert ranked.size == 2  # top 50% of 4
    assert e.n_fits_ranked == 2
    # elites are fits 5 (idx1 pos) and 6 (idx1 neg) -> noise inds re-indexed mod 2
    np.testing.assert_array_equal(np.sort(e.noise_inds), [200, 200])
