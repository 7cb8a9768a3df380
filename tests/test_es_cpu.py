"""End-to-end episodic ES on CPU, single process (the reference's
simple_example.py flow) — coverage the reference lacked."""
import numpy as np
import torch

from es_pytorch_amd.core import es
from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout import RewardResult, run_model
from es_pytorch_amd.utils.rankers import CenteredRanker
from es_pytorch_amd.utils.reporters import StdoutReporter


def test_es_step_cartpole():
    torch.manual_seed(0)
    comm = Comm(torch.device("cpu"))
    cfg = AttrDict({
        "general": {"policies_per_gen": 8, "batch_size": 100},
        "policy": {"l2coeff": 0.005},
    })
    env = make("CartPole-v1")
    env.seed(0)
    rs = np.random.RandomState(1)
    nn = FeedForward([8], torch.nn.Tanh(), env, ac_std=0.01, ob_clip=5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.05))
    nt = NoiseTable(len(policy), NoiseTable.make_noise(100_000, seed=2))
    ranker = CenteredRanker()

    def fit_fn(model, use_noise=True):
        rews, behv, obs, steps = run_model(model, env, 200, rs if use_noise else None)
        return RewardResult(rews, behv, obs, steps)

    flat_before = policy.flat_params.copy()
    reporter = StdoutReporter(comm)
    for _ in range(2):
        tr, gen_obstat = es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, reporter)
        policy.update_obstat(gen_obstat)
    assert not np.array_equal(policy.flat_params, flat_before)  # params moved
    assert policy.obstat.count > 0
    assert ranker.n_fits_ranked == 8
    assert tr.steps > 0


def test_test_params_shapes():
    torch.manual_seed(0)
    comm = Comm(torch.device("cpu"))
    env = make("Pendulum-v1")
    rs = np.random.RandomState(1)
    nn = FeedForward([8], torch.nn.Tanh(), env, ac_std=0.0, ob_clip=5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.05))
    nt = NoiseTable(len(policy), NoiseTable.make_noise(50_000, seed=2))
    obstat = ObStat(env.observation_space.shape, 0)

    def fit_fn(model):
        rews, behv, obs, steps = run_model(model, env, 50, rs)
        return RewardResult(rews, behv, obs, steps)

    pos, neg, inds, steps = es.test_params(comm, 3, policy, nt, obstat, fit_fn, rs)
    assert pos.shape == (3, 1) and neg.shape == (3, 1) and inds.shape == (3,)
    # run_model reports the last step INDEX (reference gym_runner.py:67 returns `step`)
    assert steps == 6 * 49
    assert obstat.count > 0
