"""RunCheckpointer (utils/checkpoint.py): atomic ring + bit-exact resume.

The headline property: N generations straight produces bitwise-identical
parameters to k generations + save + restore-into-fresh-objects + N-k
generations — params, optimizer moments, ObStat, RNG streams, decay
schedules and the env episode-seed counter all survive the round trip.
"""
import os

import numpy as np
import pytest
import torch

from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.core import es
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout import RewardResult, run_model
from es_pytorch_amd.utils.checkpoint import RunCheckpointer
from es_pytorch_amd.utils.rankers import CenteredRanker
from es_pytorch_amd.utils.reporters import StdoutReporter


def _fresh():
    torch.manual_seed(3)
    comm = Comm(torch.device("cpu"))
    cfg = AttrDict({"general": {"policies_per_gen": 8, "batch_size": 100},
                    "policy": {"l2coeff": 0.005},
                    "noise": {"std": 0.05, "std_decay": 0.99, "std_limit": 0.01}})
    env = make("CartPole-v1")
    env.seed(0)
    rs = np.random.RandomState(7)
    nn = FeedForward([8], torch.nn.Tanh(), env, ac_std=0.01, ob_clip=5)
    policy = Policy(nn, cfg.noise.std, Adam(len(Policy.get_flat(nn)), 0.05))
    nt = NoiseTable(len(policy), NoiseTable.make_noise(100_000, seed=2))
    return comm, cfg, env, rs, policy, nt, CenteredRanker()


def _run(objs, n):
    comm, cfg, env, rs, policy, nt, ranker = objs
    reporter = StdoutReporter(comm)

    def fit_fn(model, use_noise=True):
        rews, behv, obs, steps = run_model(model, env, 80, rs if use_noise else None)
        return RewardResult(rews, behv, obs, steps)

    for _ in range(n):
        tr, gen_obstat = es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker,
                                 reporter)
        policy.update_obstat(gen_obstat)
        # schedule mutation like obj.py (must survive the resume too)
        cfg.noise.std = policy.std = max(cfg.noise.std * cfg.noise.std_decay,
                                         cfg.noise.std_limit)


def test_exact_resume(tmp_path):
    a = _fresh()
    _run(a, 6)

    b = _fresh()
    _run(b, 3)
    comm, cfg, env, rs, policy, nt, _ = b
    ck = RunCheckpointer(str(tmp_path / "ring"), comm, keep=2)
    ck.save(3, policy, rs, cfg=cfg, env=env)

    c = _fresh()
    comm2, cfg2, env2, rs2, policy2, nt2, _ = c
    ck2 = RunCheckpointer(str(tmp_path / "ring"), comm2, keep=2)
    state = ck2.load()
    assert state is not None
    next_gen, extra = ck2.restore(state, policy2, rs2, cfg=cfg2, env=env2)
    assert next_gen == 3
    _run(c, 3)

    ap, cp = a[4], policy2
    np.testing.assert_array_equal(ap.flat_params, cp.flat_params)
    assert ap.std == cp.std and ap.optim.t == cp.optim.t
    np.testing.assert_array_equal(ap.optim.m, cp.optim.m)
    assert ap.obstat.count == cp.obstat.count
    np.testing.assert_array_equal(np.asarray(ap.obstat.mean),
                                  np.asarray(cp.obstat.mean))


def test_ring_prune_and_atomicity(tmp_path):
    comm, cfg, env, rs, policy, nt, _ = _fresh()
    ck = RunCheckpointer(str(tmp_path / "ring"), comm, keep=2, every=2)
    saved = [g for g in range(1, 7) if ck.maybe_save(g, policy, rs)]
    assert saved == [2, 4, 6]  # cadence respected
    names = sorted(os.listdir(tmp_path / "ring"))
    assert names == ["ckpt-4.pkl", "ckpt-6.pkl"]  # ring pruned to keep=2
    assert not any(n.endswith(".tmp") for n in names)  # atomic writes
    assert ck.latest().endswith("ckpt-6.pkl")


def test_world_size_mismatch_rejected(tmp_path):
    comm, cfg, env, rs, policy, nt, _ = _fresh()
    ck = RunCheckpointer(str(tmp_path / "ring"), comm)
    ck.save(1, policy, rs)
    state = ck.load()
    state["world_size"] = 8
    with pytest.raises(RuntimeError, match="world_size"):
        ck.restore(state, policy, rs)


def _ckpt_roundtrip_rank(rank, world, folder):
    import torch as T

    from es_pytorch_amd.nn.optimizers import Adam as A2
    from es_pytorch_amd.parallel.comm import Comm as C2

    T.manual_seed(1)
    env = make("CartPole-v1")
    nn = FeedForward([4], T.nn.Tanh(), env, ac_std=0.0, ob_clip=5)
    policy = Policy(nn, 0.02, A2(len(Policy.get_flat(nn)), 0.01))
    comm = C2(T.device("cpu"))
    rs = np.random.RandomState(100 + rank)  # distinct per-rank stream
    rs.randint(0, 1000, size=5)
    policy.flat_params[:] = rank + 1.0

    ck = RunCheckpointer(folder, comm, keep=2)
    ck.save(7, policy, rs)
    expected = rs.randint(0, 10 ** 6, size=8)  # what the live run draws next

    policy.flat_params[:] = -1.0  # dirty the state
    rs2 = np.random.RandomState(0)
    ck2 = RunCheckpointer(folder, comm, keep=2)
    next_gen, _ = ck2.restore(ck2.load(), policy, rs2)
    assert next_gen == 7
    # rank 0 wrote the file; every rank restores ITS OWN rng stream slice
    got = rs2.randint(0, 10 ** 6, size=8)
    np.testing.assert_array_equal(got, expected)
    # policy comes from rank 0's copy (updates are rank-identical by design)
    np.testing.assert_array_equal(policy.flat_params,
                                  np.full_like(policy.flat_params, 1.0))
    return int(expected[0])


def test_checkpoint_world2(tmp_path):
    from tests.mp_helpers import run_mp
    draws = run_mp(_ckpt_roundtrip_rank, world=2, args=(str(tmp_path / "ring"),))
    assert draws[0] != draws[1]  # streams really are per-rank


def _ckpt_save_only_rank(rank, world, folder):
    import torch as T

    from es_pytorch_amd.nn.optimizers import Adam as A2
    from es_pytorch_amd.parallel.comm import Comm as C2

    T.manual_seed(1)
    env = make("CartPole-v1")
    nn = FeedForward([8], T.nn.Tanh(), env, ac_std=0.01, ob_clip=5)  # _fresh arch
    policy = Policy(nn, 0.02, A2(len(Policy.get_flat(nn)), 0.01))
    policy.flat_params[:] = 42.0
    rs = np.random.RandomState(100 + rank)
    RunCheckpointer(folder, C2(T.device("cpu")), keep=2).save(
        9, policy, rs, extra={"tag": "reshard"})
    return rank


def test_elastic_reshard_restore(tmp_path):
    """A snapshot written at world_size=2 restores into a 1-rank run with
    allow_reshard=True: learned state carries over exactly, per-rank RNG
    streams re-split deterministically; strict mode still refuses."""
    from tests.mp_helpers import run_mp
    folder = str(tmp_path / "ring")
    run_mp(_ckpt_save_only_rank, world=2, args=(folder,))

    comm, cfg, env, rs, policy, nt, _ = _fresh()  # world_size=1 here
    ck = RunCheckpointer(folder, comm)
    state = ck.load()
    with pytest.raises(RuntimeError, match="allow_reshard"):
        ck.restore(state, policy, rs)
    next_gen, extra = ck.restore(state, policy, rs, allow_reshard=True)
    assert next_gen == 9 and extra["tag"] == "reshard"
    np.testing.assert_array_equal(policy.flat_params,
                                  np.full_like(policy.flat_params, 42.0))
    a = rs.randint(0, 10 ** 6, 4)
    # deterministic: a second resharded restore reproduces the same stream
    rs2 = np.random.RandomState(1)
    ck.restore(state, policy, rs2, allow_reshard=True)
    np.testing.assert_array_equal(rs2.randint(0, 10 ** 6, 4), a)
