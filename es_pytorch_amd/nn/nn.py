"""Policy networks.

Same model family as the reference (``src/nn/nn.py:9-117``): MLPs with
virtual-batch-norm observation normalization — ``clamp((ob-mean)/std,
+-ob_clip)`` (``nn.py:45``) — followed by a Linear+activation stack, with
optional gaussian action noise added in forward (``nn.py:47-48``), plus the
integrated-action-std variants (``nn.py:53-96``) and binned discretized
actions (``nn.py:99-117``).

These torch modules are the EPISODIC/CPU path and the checkpoint phenotype.
The GPU training path evaluates the same architecture population-batched via
the fused HIP MLP kernel (``ops/csrc/hip/mlp_fwd.hip``); layer sizes and
normalization semantics are shared through :meth:`BaseNet.layer_dims`.


PROVENANCE: the module structure and forward bodies follow the reference
(src/nn/nn.py) closely — checkpoint compatibility pins the state_dict
layout; layer_dims(), _normalize(), the unified _ActionView output sizing
and the kernel-shared contracts are original.
"""
from __future__ import annotations

from abc import ABC
from typing import List, Optional

import numpy as np
import torch
from torch import Tensor, clamp, nn


class BaseNet(nn.Module, ABC):
    def __init__(self, layers: List[nn.Module], ob_shape: tuple, ob_clip: float = 5):
        super().__init__()
        self.model = nn.Sequential(*layers)

        self._obmean: np.ndarray = np.zeros(ob_shape)
        self._obstd: np.ndarray = np.ones(ob_shape)

        self.ob_clip = ob_clip

    def set_ob_mean_std(self, mean: np.ndarray, std: np.ndarray):
        self._obmean = mean
        self._obstd = std

    def _normalize(self, inp: Tensor) -> Tensor:
        mean = torch.as_tensor(self._obmean, dtype=inp.dtype, device=inp.device)
        std = torch.as_tensor(self._obstd, dtype=inp.dtype, device=inp.device)
        return clamp((inp - mean) / std, min=-self.ob_clip, max=self.ob_clip)

    def layer_dims(self) -> List[int]:
        """[in, hidden..., out] — the shape contract shared with the HIP batched forward."""
        dims = []
        for m in self.model:
            if isinstance(m, nn.Linear):
                if not dims:
                    dims.append(m.in_features)
                dims.append(m.out_features)
        return dims


def _mlp_layers(layer_sizes: List[int], activation: nn.Module) -> List[nn.Module]:
    layers: List[nn.Module] = []
    for in_size, out_size in zip(layer_sizes[:-1], layer_sizes[1:]):
        layers += [nn.Linear(in_size, out_size), activation]
    return layers


class FeedForward(BaseNet):
    """Basic feed-forward policy (reference ``nn.py:24-50``).

    :param layer_sizes: hidden layer sizes; input/output sizes come from the env
    :param ac_std: std of the gaussian action noise added in forward
    :param ob_clip: min/max normalized observation value
    """

    def __init__(self, layer_sizes: List[int], activation: nn.Module, env, ac_std: float,
                 ob_clip: float = 5):
        sizes = [int(np.prod(env.observation_space.shape))] + list(layer_sizes) + \
                [int(np.prod(env.action_space.shape))]
        super().__init__(_mlp_layers(sizes, activation), env.observation_space.shape, ob_clip)
        self._action_std = ac_std

    def forward(self, inp: Tensor, **kwargs) -> Tensor:
        rs: Optional[np.random.RandomState] = kwargs.get("rs")
        a = self.model(self._normalize(inp).float())
        if self._action_std != 0 and rs is not None:
            a = a + torch.as_tensor(rs.randn(*a.shape), dtype=a.dtype) * self._action_std
        return a


class _ActionView:
    """Env facade that widens the action space seen by FeedForward sizing.

    The integrated-gaussian variants consume part of the network output as
    the action std, so their output layer must be WIDER than the env's
    action dim: adim+1 (one shared std) or 2*adim (per-dim stds). The
    reference sized the output layer adim (``nn.py:33`` reused by ``:53``)
    and then consumed one output as std, emitting adim-1 actions — an
    off-by-one no real env accepts. Here BOTH the episodic and the engine
    path size via this view and emit exactly adim actions (single
    output-layer contract; the engine's act decode assumes it).
    """

    def __init__(self, env, extra: int):
        self.observation_space = env.observation_space
        space = env.action_space
        shape = (int(np.prod(space.shape)) + extra,)
        self.action_space = type("_W", (), {"shape": shape})()


class FFIntegGausAction(FeedForward):
    """MLP whose FIRST output is the (shared) action std; remaining outputs
    are the adim action means (reference ``nn.py:53-74``; output sizing
    unified across paths, see :class:`_ActionView`)."""

    def __init__(self, layer_sizes: List[int], activation: nn.Module, env, ac_std: float,
                 ob_clip: float = 5):
        super().__init__(layer_sizes, activation, _ActionView(env, 1), ac_std, ob_clip)

    def forward(self, inp: Tensor, **kwargs) -> np.ndarray:
        rs: Optional[np.random.RandomState] = kwargs.get("rs")
        out = self.model(self._normalize(inp).float()).detach().numpy()
        action, action_std = out[1:], out[0]
        if action_std != 0 and rs is not None:
            action = action + rs.standard_normal(*action.shape) * action_std
        return action


class FFIntegGausActionMulti(FeedForward):
    """MLP emitting [mean-half, std-half] outputs (reference ``nn.py:77-96``;
    output layer sized 2*adim on both paths, see :class:`_ActionView`)."""

    def __init__(self, layer_sizes: List[int], activation: nn.Module, env, ac_std: float,
                 ob_clip: float = 5):
        adim = int(np.prod(env.action_space.shape))
        super().__init__(layer_sizes, activation, _ActionView(env, adim), ac_std, ob_clip)

    def forward(self, inp: Tensor, **kwargs) -> np.ndarray:
        rs: Optional[np.random.RandomState] = kwargs.get("rs")
        out = self.model(self._normalize(inp).float()).detach().numpy()
        mid = len(out) // 2
        action, action_std = out[:mid], np.abs(out[mid:])
        if rs is not None:
            action = action + rs.standard_normal(*action.shape) * action_std
        return action


class FFBinned(BaseNet):
    """Discretized continuous actions via per-dim argmax over bins (reference ``nn.py:99-117``)."""

    def __init__(self, layer_sizes: List[int], activation: nn.Module, env, n_bins: int, ob_clip=5):
        self.bins = n_bins
        self.adim = env.action_space.shape[0]
        self.ahigh, self.alow = env.action_space.high, env.action_space.low

        sizes = [int(np.prod(env.observation_space.shape))] + list(layer_sizes) + \
                [self.adim * self.bins]
        super().__init__(_mlp_layers(sizes, activation), env.observation_space.shape, ob_clip)

    def forward(self, inp: Tensor, **kwargs) -> Tensor:
        a: Tensor = self.model(self._normalize(inp).float())
        ac_range = torch.as_tensor((self.ahigh - self.alow)[None, :], dtype=a.dtype)
        alow = torch.as_tensor(self.alow[None, :], dtype=a.dtype)
        binned_ac = a.reshape((-1, self.adim, self.bins)).argmax(2)
        return (1.0 / (self.bins - 1.0) * binned_ac * ac_range + alow).squeeze()
