"""Multi-process (gloo, world_size=2) correctness: the reference's parallel
test pattern — rank-dependent inputs, rank-independent expected outputs,
asserted on EVERY rank (reference test/es/es_runner_test.py,
test/utils/obstat_test.py)."""
import numpy as np
import pytest
import torch

from tests.mp_helpers import run_mp


def _share_results_worker(rank, world):
    from es_pytorch_amd.core.es import _share_results
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    # 2 evals per rank, 4 objectives (reference es_runner_test.py:10-31)
    E, O = 2, 4
    fits_pos = [[float(rank * 100 + e * 10 + o) for o in range(O)] for e in range(E)]
    fits_neg = [[float(-(rank * 100 + e * 10 + o)) for o in range(O)] for e in range(E)]
    inds = [rank * 1000 + e for e in range(E)]
    res = _share_results(comm, fits_pos, fits_neg, inds)
    assert res.shape == (world * E, 2 * O + 1)
    for r in range(world):
        for e in range(E):
            row = res[r * E + e]
            np.testing.assert_allclose(row[:O], [r * 100 + e * 10 + o for o in range(O)])
            np.testing.assert_allclose(row[O:2 * O], [-(r * 100 + e * 10 + o) for o in range(O)])
            assert row[-1] == r * 1000 + e
    return True


def test_share_results_mp():
    assert all(run_mp(_share_results_worker, world=2))


def _obstat_worker(rank, world):
    from es_pytorch_amd.nn.obstat import ObStat
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    s = ObStat((3,), 0)
    # rank-scaled inputs (reference obstat_test.py:8-23)
    s.inc(np.full(3, float(rank + 1)), np.full(3, float((rank + 1) ** 2)), rank + 1)
    s.dist_inc(comm)
    tot = sum(r + 1 for r in range(world))
    totsq = sum((r + 1) ** 2 for r in range(world))
    np.testing.assert_allclose(s.sum, np.full(3, float(tot)))
    np.testing.assert_allclose(s.sumsq, np.full(3, float(totsq)))
    assert s.count == tot
    return True


def test_obstat_dist_inc_mp():
    assert all(run_mp(_obstat_worker, world=2))


def _noisetable_worker(rank, world):
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    nt = NoiseTable.create_shared(comm, 10_000, 10, seed=None)  # rank0 draws, broadcasts
    # all ranks must hold the identical table (reference noisetable_test.py:19-26)
    h = float(nt.noise.sum())
    hs = comm.allgather_obj(h)
    assert all(abs(x - hs[0]) < 1e-6 for x in hs)
    return True


def test_noisetable_shared_seed_mp():
    assert all(run_mp(_noisetable_worker, world=2))


def _es_e2e_worker(rank, world):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core import es
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm, seed_all
    from es_pytorch_amd.rollout import RewardResult, run_model
    from es_pytorch_amd.utils.rankers import CenteredRanker
    from es_pytorch_amd.utils.reporters import StdoutReporter

    comm = Comm(torch.device("cpu"))
    rs, my_seed, global_seed = seed_all(comm, [11, 22][:world])
    env = make("CartPole-v1")
    env.seed(my_seed)
    cfg = AttrDict({"general": {"policies_per_gen": 8, "batch_size": 100},
                    "policy": {"l2coeff": 0.005}})
    nn = FeedForward([8], torch.nn.Tanh(), env, ac_std=0.01, ob_clip=5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.05))
    nt = NoiseTable.create_shared(comm, 100_000, len(policy), seed=3)
    ranker = CenteredRanker()

    def fit_fn(model, use_noise=True):
        rews, behv, obs, steps = run_model(model, env, 100, rs if use_noise else None)
        return RewardResult(rews, behv, obs, steps)

    for _ in range(2):
        es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, StdoutReporter(comm))

    # the load-bearing invariant: every rank computed the IDENTICAL update
    # redundantly (reference README.md:10-12 design), so flat params match.
    checks = comm.allgather_obj(float(np.abs(policy.flat_params).sum()))
    assert all(abs(c - checks[0]) < 1e-5 for c in checks), checks
    return True


@pytest.mark.timeout(300)
def test_es_end_to_end_mp():
    assert all(run_mp(_es_e2e_worker, world=2, timeout=280))


def _broadcast_tensor_worker(rank, world):
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    # rank-dependent input, rank-0's value expected everywhere
    t = torch.tensor([float(rank + 1), float(10 * rank)], dtype=torch.float64)
    comm.broadcast_tensor_(t, src=0)
    np.testing.assert_allclose(t.numpy(), [1.0, 0.0])
    return True


def test_broadcast_tensor_mp():
    """Comm.broadcast_tensor_ (the archive-growth primitive) across 2 ranks."""
    assert all(run_mp(_broadcast_tensor_worker, world=2))


def test_comm_single_process_fallbacks():
    """With no initialized process group, every collective must degrade to
    a correct single-process identity (the world_size=1 path every
    single-GPU run takes)."""
    from es_pytorch_amd.parallel.comm import Comm
    comm = Comm(torch.device("cpu"))
    assert comm.size == 1 and comm.rank == 0

    rows = torch.arange(6, dtype=torch.float64).reshape(2, 3)
    assert torch.equal(comm.allgather_rows(rows), rows)

    t = torch.ones(3)
    comm.allreduce_sum_(t)
    assert torch.equal(t, torch.ones(3))

    assert comm.allreduce_scalar(2.5) == 2.5
    assert comm.broadcast_obj({"a": 1}) == {"a": 1}
    b = torch.tensor([4.0, 5.0])
    comm.broadcast_tensor_(b)
    assert torch.equal(b, torch.tensor([4.0, 5.0]))
    assert comm.allgather_obj("x") == ["x"]
    comm.barrier()  # no-op, must not hang
