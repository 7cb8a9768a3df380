"""Pair-episode rollout (one launch per generation / k-step chunks) must be
bitwise-identical to per-step pair launches: same body, same salt sequence,
only the launch geometry differs (rollout_loco.hip pair-episode kernels)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda", 0)


def _run(dev, rollout_mode, chunk=1, pop=128, steps=30):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(11)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": steps},
                    "noise": {"tbl_size": 2_000_000, "std": 0.02},
                    "policy": {"layer_sizes": [64, 64], "ac_std": 0.01,
                               "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                               "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": pop, "batch_size": 500,
                                "seed": 3, "steps_per_launch": chunk}})
    env = make_batched("Humanoid-v2", pop + 1, dev, max_steps=steps,
                       terminate_on_fall=True)
    nn = FeedForward([64, 64], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 2_000_000, len(policy), seed=7, device=dev)
    rs = np.random.RandomState(5)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False,
                    rollout_mode=rollout_mode, pair_rollout=True)
    assert eng.pair_rollout
    ranker = CenteredRanker()
    eng.step(ranker)
    torch.cuda.synchronize(dev)
    return (np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel(),
            eng.rew_total.cpu().numpy().copy(),
            eng.behv.cpu().numpy().copy(),
            eng.member_steps.cpu().numpy().copy(),
            eng.theta.cpu().numpy().copy())


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
@pytest.mark.parametrize("mode,chunk", [("episode", 1), ("step", 8)])
def test_pair_episode_bitwise_matches_step(dev, mode, chunk):
    ref = _run(dev, "step", chunk=1)
    out = _run(dev, mode, chunk=chunk)
    for a, b, name in zip(ref, out, ["fits", "rew_total", "behv", "steps", "theta"]):
        np.testing.assert_array_equal(a, b, err_msg=f"{mode}/chunk{chunk}:{name}")
