"""On-device NSR-A archive growth (engine.grow_archive)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda", 0)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_grow_archive_appends_noiseless_behaviour(dev):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.novelty import novelty
    from es_pytorch_amd.utils.rankers import CenteredRanker, MultiObjectiveRanker

    torch.manual_seed(4)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 20},
                    "noise": {"tbl_size": 1_000_000, "std": 0.02},
                    "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 16, "batch_size": 500, "seed": 2}})
    env = make_batched("Humanoid-v2", 17, dev, max_steps=20, terminate_on_fall=False)
    nn = FeedForward([32], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=6, device=dev)
    eng = GpuEngine(cfg, comm, policy, nt, env, np.random.RandomState(1),
                    objective="nsr", novelty_k=3, use_graph=False)
    eng.archive = torch.randn(8, 2, dtype=torch.float64, device=dev,
                              generator=None) * 3.0

    before = eng.archive.cpu().numpy().copy()
    eng.step(MultiObjectiveRanker(CenteredRanker(), 0.5))
    nov = eng.grow_archive()
    after = eng.archive.cpu().numpy()

    assert after.shape == (9, 2)
    np.testing.assert_array_equal(after[:8], before)
    # the appended row is the noiseless slot's final (x, y)
    nl = eng._member_behv()[-1].cpu().numpy()
    np.testing.assert_allclose(after[-1], nl[:2], rtol=0, atol=0)
    # novelty scored BEFORE the append (reference nsra.py:130-133 order),
    # identical to the host reference implementation
    host = novelty(after[-1], before, 3)
    np.testing.assert_allclose(nov, host, rtol=1e-6)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_noiseless_eval_mutates_nothing(dev):
    """The archive-init evaluation: no optimizer update, no
    numpy RNG consumption, params untouched — unlike a full step()."""
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm

    torch.manual_seed(12)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 15},
                    "noise": {"tbl_size": 1_000_000, "std": 0.02},
                    "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 4}})
    env = make_batched("Humanoid-v2", 9, dev, max_steps=15, terminate_on_fall=False)
    nn = FeedForward([32], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=2, device=dev)
    rs = np.random.RandomState(9)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False)

    rs_state = rs.get_state()[1].copy()
    theta0 = eng.theta.cpu().numpy().copy()
    t0, m0 = policy.optim.t, policy.optim.m.copy()

    rew, behv, steps = eng.noiseless_eval()

    assert np.isfinite(rew) and steps == 15 * eng.eps and behv.shape == (3,)
    np.testing.assert_array_equal(eng.theta.cpu().numpy(), theta0)
    assert policy.optim.t == t0
    np.testing.assert_array_equal(policy.optim.m, m0)
    np.testing.assert_array_equal(rs.get_state()[1], rs_state)
