"""Property-based tests (hypothesis) for the numeric substrate.

The reference's tests pin a handful of hand-picked fixtures;
these properties pin the ALGEBRA the ES update relies on, across generated
inputs: rank-transform invariances, antithetic symmetry, ObStat merge
associativity/commutativity, and the forward<->flat layout permutation
being a true bijection.
"""
import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st
from hypothesis.extra.numpy import arrays

from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.utils.rankers import CenteredRanker, rank

finite_floats = st.floats(min_value=-1e6, max_value=1e6, allow_nan=False,
                          width=32)


@given(arrays(np.float64, st.integers(2, 64), elements=finite_floats))
@settings(max_examples=50, deadline=None)
def test_rank_is_a_permutation_respecting_order(x):
    r = rank(x)
    assert sorted(r) == list(range(len(x)))          # a permutation of 0..n-1
    order = np.argsort(x, kind="stable")
    assert (r[order] == np.arange(len(x))).all()     # consistent with sorting


@given(arrays(np.float64, st.integers(2, 64),
              elements=st.integers(-10 ** 6, 10 ** 6).map(float)),
       st.sampled_from([0.5, 1.0, 2.0, 1024.0]))
@settings(max_examples=50, deadline=None)
def test_centered_rank_is_scale_and_shift_invariant(x, scale):
    """The OpenAI-ES shaping must depend only on the ORDER of fitnesses.
    (Scale is a power of two and values sit on an integer grid so the affine
    transform is order-exact — hypothesis found that e.g. 3e-41 + 3.7
    absorbs to 3.7 and legitimately changes ties.)"""
    cr = CenteredRanker()
    a = cr._rank(x.copy())
    b = cr._rank((x * scale + 3.0).copy())
    np.testing.assert_array_equal(a, b)
    assert a.min() >= -0.5 - 1e-6 and a.max() <= 0.5 + 1e-6


@given(arrays(np.float64, st.integers(1, 32), elements=finite_floats),
       st.integers(0, 2 ** 31))
@settings(max_examples=50, deadline=None)
def test_antithetic_post_rank_antisymmetry(fits, seed):
    """Swapping the + and - halves must negate the shaped fitnesses: the
    antithetic difference ranked[:n] - ranked[n:] is what makes the pair
    trick variance-reducing."""
    rng = np.random.RandomState(seed % (2 ** 31))
    neg = rng.randn(len(fits))
    inds = np.arange(len(fits))
    a = CenteredRanker().rank(fits.reshape(-1, 1), neg.reshape(-1, 1), inds)
    b = CenteredRanker().rank(neg.reshape(-1, 1), fits.reshape(-1, 1), inds)
    np.testing.assert_allclose(a, -b, atol=1e-12)


@given(st.lists(st.tuples(
    arrays(np.float64, 4, elements=finite_floats),
    st.floats(min_value=0, max_value=1e3)), min_size=2, max_size=6))
@settings(max_examples=40, deadline=None)
def test_obstat_merge_is_order_independent(parts):
    """inc/__iadd__ must commute and associate: rank order in the packed
    all_reduce must not change the merged statistics."""
    def merged(order):
        s = ObStat((4,), 0)
        for i in order:
            o = ObStat((4,), 0)
            vals, cnt = parts[i]
            o.inc(vals, np.abs(vals), cnt)
            s += o
        return s

    fwd = merged(range(len(parts)))
    rev = merged(reversed(range(len(parts))))
    # fp addition is order-sensitive at the last ulps; the merge contract is
    # mathematical order-independence, asserted to ~1e-9 relative
    np.testing.assert_allclose(fwd.sum, rev.sum, rtol=1e-9, atol=1e-6)
    np.testing.assert_allclose(fwd.sumsq, rev.sumsq, rtol=1e-9, atol=1e-6)
    np.testing.assert_allclose(fwd.count, rev.count, rtol=1e-12, atol=0)


@given(st.lists(st.integers(1, 40), min_size=2, max_size=5))
@settings(max_examples=40, deadline=None)
def test_forward_perm_is_a_bijection(dims):
    """The engine's forward-layout permutation must be a true bijection of
    [0, n): every flat parameter maps to exactly one forward slot."""
    from es_pytorch_amd.core.engine import forward_perm
    perm = forward_perm(dims).numpy()
    n = sum(i * o + o for i, o in zip(dims[:-1], dims[1:]))
    assert perm.shape == (n,)
    assert np.array_equal(np.sort(perm), np.arange(n))


@given(st.integers(0, 2 ** 31 - 1), st.integers(1, 3))
@settings(max_examples=20, deadline=None)
def test_noise_table_slices_match_full_fill(seed, k):
    """NoiseTable slices are views of one deterministic stream: re-creating
    the table from the same seed reproduces any slice bitwise."""
    import torch

    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    a = NoiseTable.create_shared(comm, 10_000, 100, seed=seed)
    b = NoiseTable.create_shared(comm, 10_000, 100, seed=seed)
    idx = (seed % 7919) % (10_000 - 100 * k)
    np.testing.assert_array_equal(a.get(idx, 100 * k).numpy(),
                                  b.get(idx, 100 * k).numpy())


@given(st.lists(st.integers(1, 33), min_size=2, max_size=4))
@settings(max_examples=60, deadline=None)
def test_fp8_interleave_map_is_a_bijection(dims):
    """The e4m3 blob layout (pheno.hip fp8_src_elem, mirrored by the test
    helper) must be a PERMUTATION of [0, n) for every layer shape — even,
    odd and scalar-path layers alike; a collision or gap would silently
    corrupt perturbations."""
    import importlib.util
    import os
    spec = importlib.util.spec_from_file_location(
        "fp8_test_helpers",
        os.path.join(os.path.dirname(__file__), "test_fp8.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    n = sum(i * o + o for i, o in zip(dims[:-1], dims[1:]))
    m = mod._interleave_map(dims, n)
    assert np.array_equal(np.sort(m), np.arange(n)), dims
