"""Policy: flat/pheno round-trip and pickle checkpoint format
(coverage the reference lacked — SURVEY.md §4)."""
import os

import numpy as np
import torch

from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.optimizers import Adam


def _policy(seed=0):
    torch.manual_seed(seed)
    env = make("CartPole-v1")
    nn = FeedForward([8, 8], torch.nn.Tanh(), env, ac_std=0.0, ob_clip=5)
    return Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01)), env


def test_flat_roundtrip():
    p, _ = _policy()
    flat = p.flat_params.copy()
    p.set_nn_params(flat)
    np.testing.assert_array_equal(Policy.get_flat(p._module), flat)


def test_pheno_applies_noise():
    p, _ = _policy()
    base = p.flat_params.copy()
    noise = np.random.RandomState(0).randn(len(p)).astype(np.float32)
    p.pheno(noise)
    np.testing.assert_allclose(Policy.get_flat(p._module), base + p.std * noise, rtol=1e-6)
    # flat_params untouched by pheno
    np.testing.assert_array_equal(p.flat_params, base)
    # zero noise restores exactly
    p.pheno()
    np.testing.assert_array_equal(Policy.get_flat(p._module), base)


def test_pheno_forward_runs():
    p, env = _policy()
    ob = torch.from_numpy(env.reset()).float()
    with torch.no_grad():
        a = p.pheno()(ob, rs=None)
    assert a.shape == (1,)


def test_save_load_roundtrip(tmp_path):
    p, _ = _policy()
    p.optim_step(np.ones(len(p), dtype=np.float32))
    p.obstat.inc(np.ones(4), np.ones(4), 5)
    p.save(str(tmp_path), "7")
    fp = os.path.join(str(tmp_path), "policy-7")
    assert os.path.exists(fp)
    q = Policy.load(fp)
    np.testing.assert_array_equal(q.flat_params, p.flat_params)
    np.testing.assert_array_equal(q.optim.m, p.optim.m)
    assert q.optim.t == p.optim.t
    assert q.obstat.count == p.obstat.count
    # module weights re-synced from flat vector on load
    np.testing.assert_array_equal(Policy.get_flat(q._module), q.flat_params)


def test_identical_seeding_gives_identical_params():
    p1, _ = _policy(seed=123)
    p2, _ = _policy(seed=123)
    np.testing.assert_array_equal(p1.flat_params, p2.flat_params)
