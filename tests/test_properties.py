"""Property-based tests (hypothesis) for the numeric substrate."""
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st
from hypothesis.extra.numpy import arrays

from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.utils.rankers import CenteredRanker, rank
from es_pytorch_amd.utils.utils import scale_noise

finite_floats = st.floats(min_value=-1e6, max_value=1e6, allow_nan=False,
                          allow_infinity=False, width=32)


@settings(max_examples=50, deadline=None)
@given(arrays(np.float64, st.integers(2, 64), elements=finite_floats, unique=True))
def test_rank_is_permutation(x):
    r = rank(x)
    assert sorted(r) == list(range(len(x)))
    # order-preserving: larger value -> larger rank
    order = np.argsort(x)
    assert list(r[order]) == list(range(len(x)))


@settings(max_examples=50, deadline=None)
@given(arrays(np.float64, st.integers(1, 32), elements=finite_floats, unique=True),
       arrays(np.float64, st.integers(1, 32), elements=finite_floats, unique=True))
def test_centered_ranker_bounds_and_antisymmetry(a, b):
    n = min(len(a), len(b))
    if n < 1 or len(set(np.concatenate([a[:n], b[:n]]).tolist())) < 2 * n:
        return
    pos, neg = a[:n].reshape(-1, 1), b[:n].reshape(-1, 1)
    inds = np.arange(n)
    r1 = CenteredRanker().rank(pos, neg, inds)
    # antithetic difference of two [-0.5, 0.5] ranks
    assert np.all(np.abs(r1) <= 1.0 + 1e-9)
    # swapping pos and neg flips the sign
    r2 = CenteredRanker().rank(neg, pos, inds)
    np.testing.assert_allclose(r1, -r2, atol=1e-6)


@settings(max_examples=30, deadline=None)
@given(st.integers(0, 10_000), st.integers(2, 40), st.integers(1, 7))
def test_scale_noise_linearity(seed, n_rows, batch):
    rs = np.random.RandomState(seed)
    nt = NoiseTable(8, torch.arange(2000, dtype=torch.float32))
    inds = rs.randint(0, 1990, size=n_rows)
    fits = rs.randn(n_rows).astype(np.float32)
    g1 = scale_noise(fits, inds, nt, 8, batch).numpy()
    g2 = scale_noise(2 * fits, inds, nt, 8, batch).numpy()
    np.testing.assert_allclose(g2, 2 * g1, rtol=1e-5, atol=1e-3)


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(finite_floats, st.integers(1, 10)), min_size=1, max_size=8))
def test_obstat_merge_associativity(chunks):
    """Incremental accumulation == one-shot accumulation."""
    inc = ObStat((2,), 0)
    tot_s = np.zeros(2)
    tot_q = np.zeros(2)
    tot_c = 0
    for v, c in chunks:
        s = np.full(2, v) * c
        q = np.full(2, v * v) * c
        inc.inc(s, q, c)
        tot_s += s
        tot_q += q
        tot_c += c
    one = ObStat((2,), 0)
    one.inc(tot_s, tot_q, tot_c)
    np.testing.assert_allclose(inc.mean, one.mean, rtol=1e-9)
    np.testing.assert_allclose(inc.std, one.std, rtol=1e-9)


@settings(max_examples=20, deadline=None)
@given(st.integers(0, 2**31), st.integers(1, 4096))
def test_noise_prefix_property(seed, n):
    """Philox element i depends only on (seed, i), never on table size."""
    a = NoiseTable.make_noise(n + 17, seed=seed)
    b = NoiseTable.make_noise(n, seed=seed)
    assert torch.equal(a[:n], b)
