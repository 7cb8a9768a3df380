"""Network family forward semantics on the episodic path
(reference src/nn/nn.py:24-117)."""
import numpy as np
import torch

from es_pytorch_amd.envs import make
from es_pytorch_amd.nn.nn import (FeedForward, FFBinned, FFIntegGausAction,
                                  FFIntegGausActionMulti)
from es_pytorch_amd.spaces import Box


class _Env:
    observation_space = Box(-np.inf, np.inf, (6,))
    action_space = Box(-1.0, 1.0, (3,))


def test_feedforward_shapes_and_clip():
    torch.manual_seed(0)
    nn = FeedForward([8, 8], torch.nn.Tanh(), _Env, ac_std=0.0, ob_clip=2.0)
    ob = torch.tensor([100.0, -100.0, 0, 0, 0, 0])
    a = nn(ob, rs=None)
    assert a.shape == (3,)
    assert a.abs().max() <= 1.0  # tanh output
    # normalization clip applied (ob 100 with mean 0 std 1 -> clipped to 2)
    nn2 = FeedForward([8], torch.nn.Identity(), _Env, 0.0, ob_clip=2.0)
    x = nn2._normalize(ob)
    assert x.max() == 2.0 and x.min() == -2.0


def test_feedforward_action_noise_used():
    torch.manual_seed(0)
    nn = FeedForward([8], torch.nn.Tanh(), _Env, ac_std=0.5, ob_clip=5)
    ob = torch.zeros(6)
    a1 = nn(ob, rs=np.random.RandomState(1))
    a2 = nn(ob, rs=np.random.RandomState(2))
    a3 = nn(ob, rs=None)
    assert not torch.allclose(a1, a2)
    assert not torch.allclose(a1, a3)


def test_integ_gaus_action():
    torch.manual_seed(0)
    nn = FFIntegGausAction([8], torch.nn.Tanh(), _Env, ac_std=0.0, ob_clip=5)
    ob = torch.zeros(6)
    a = nn(ob, rs=np.random.RandomState(0))
    # unified contract: output layer sized adim+1 (first output = std),
    # so the env's full adim actions come out — the reference's sizing
    # (nn.py:33 reused by :53) emitted adim-1, which no env accepts
    assert nn.layer_dims()[-1] == 4
    assert a.shape == (3,)


def test_integ_gaus_action_multi():
    torch.manual_seed(0)

    class _Env4:
        observation_space = Box(-np.inf, np.inf, (6,))
        action_space = Box(-1.0, 1.0, (4,))

    nn = FFIntegGausActionMulti([8], torch.nn.Tanh(), _Env4, ac_std=0.0, ob_clip=5)
    a = nn(torch.zeros(6), rs=np.random.RandomState(0))
    assert nn.layer_dims()[-1] == 8  # [mean-half | std-half]
    assert a.shape == (4,)


def test_integ_gaus_sizing_matches_engine_contract():
    """Cross-path parity: the SAME env must produce the SAME network on the
    episodic path and the engine path (engine act decode modes 2/3 expect
    adim+1 / 2*adim outputs; core/engine.py output-contract check)."""
    for cls, extra in ((FFIntegGausAction, 1), (FFIntegGausActionMulti, 3)):
        nn = cls([8], torch.nn.Tanh(), _Env, ac_std=0.0, ob_clip=5)
        dims = nn.layer_dims()
        assert dims == [6, 8, 3 + extra]


def test_binned_decode():
    torch.manual_seed(0)
    nn = FFBinned([8], torch.nn.Tanh(), _Env, n_bins=5, ob_clip=5)
    a = nn(torch.zeros(6))
    assert a.shape == (3,)
    # decoded values land on the bin lattice within [low, high]
    lattice = np.linspace(-1, 1, 5)
    for v in a.numpy():
        assert np.min(np.abs(lattice - v)) < 1e-6


def test_layer_dims_contract():
    nn = FeedForward([32, 16], torch.nn.Tanh(), _Env, 0.0, 5)
    assert nn.layer_dims() == [6, 32, 16, 3]
    env = make("Humanoid-v2")
    nn2 = FeedForward([256, 256], torch.nn.Tanh(), env, 0.0, 5)
    assert nn2.layer_dims() == [376, 256, 256, 17]
