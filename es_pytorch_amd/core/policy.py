"""Policy: the canonical flat parameter vector + its torch phenotype.

Same responsibilities and checkpoint format as the reference
(``src/core/policy.py:19-74``): owns the flat float32 vector (concatenation of
``state_dict`` tensors in order, ``policy.py:33-35``), materializes
theta + std*noise back into the torch module (``pheno``, ``policy.py:61-67``),
kaiming-normal init (``policy.py:14-16``), and whole-object pickle
checkpoints ``policy-<suffix>`` (``policy.py:37-47``) carrying params, ObStat
and optimizer state.

The GPU engine (``core/engine.py``) treats this object as the host-side truth
it syncs with once per generation; perturbation there is the batched HIP
pheno kernel, not this per-module path.


PROVENANCE: the attribute layout, flat-vector semantics and pickle format
are pinned by the reference checkpoint contract (src/core/policy.py) and
parts of this file are a direct port of those definitions; additions
(tensor-noise handling, compat unpickler for reference checkpoints, safe
file handling) are original.
"""
from __future__ import annotations

import os
import pickle

import numpy as np
import torch

from es_pytorch_amd.nn.nn import BaseNet
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.nn.optimizers import Optimizer


def init_normal(m):
    if type(m) == torch.nn.Linear:
        torch.nn.init.kaiming_normal_(m.weight)


#: reference module path -> this package (attribute-compatible classes)
_REF_MODULE_MAP = {
    "src.core.policy": "es_pytorch_amd.core.policy",
    "src.nn.nn": "es_pytorch_amd.nn.nn",
    "src.nn.obstat": "es_pytorch_amd.nn.obstat",
    "src.nn.optimizers": "es_pytorch_amd.nn.optimizers",
    "src.gym.training_result": "es_pytorch_amd.rollout.results",
}


class _CompatUnpickler(pickle.Unpickler):
    """Unpickler that accepts the reference's module paths (cross-load)."""

    def find_class(self, module: str, name: str):
        return super().find_class(_REF_MODULE_MAP.get(module, module), name)


class Policy:
    def __init__(self, module: BaseNet, noise_std: float, optim: Optimizer):
        module.apply(init_normal)

        self._module: BaseNet = module
        self.std = noise_std

        self.flat_params: np.ndarray = Policy.get_flat(module)
        self.obstat: ObStat = ObStat(module._obmean.shape, 1e-2)
        self.optim = optim

    def __len__(self):
        return len(self.flat_params)

    @staticmethod
    def get_flat(module: torch.nn.Module) -> np.ndarray:
        return torch.cat([t.flatten() for t in module.state_dict().values()]).numpy()

    @staticmethod
    def load(file: str) -> "Policy":
        """Open a checkpoint written by this repo OR by the reference.

        A pickle produced by the reference's ``Policy.save``
        (``src/core/policy.py:43-47``) stores classes under the reference's
        module paths (``src.core.policy.Policy``, ``src.nn.nn.FeedForward``,
        ...); :class:`_CompatUnpickler` remaps those onto this package's
        attribute-compatible classes, so reference checkpoints load
        byte-for-byte with no conversion step.
        """
        with open(file, "rb") as f:
            policy: Policy = _CompatUnpickler(f).load()
        policy.set_nn_params(policy.flat_params)
        return policy

    def save(self, folder: str, suffix: str):
        if not os.path.exists(folder):
            os.makedirs(folder)
        with open(os.path.join(folder, f"policy-{suffix}"), "wb") as f:
            pickle.dump(self, f)

    def set_nn_params(self, params: np.ndarray) -> torch.nn.Module:
        with torch.no_grad():
            d = {}
            curr = 0
            for name, weights in self._module.state_dict().items():
                n = weights.numel()
                d[name] = torch.from_numpy(
                    np.reshape(np.asarray(params[curr:curr + n]), weights.shape).copy())
                curr += n
            self._module.load_state_dict(d)
        return self._module

    def pheno(self, noise=None) -> torch.nn.Module:
        if noise is None:
            noise = np.zeros(len(self), dtype=np.float32)
        if isinstance(noise, torch.Tensor):
            noise = noise.cpu().numpy()
        params = self.flat_params + self.std * noise
        self.set_nn_params(params)
        return self._module

    def update_obstat(self, obstat: ObStat):
        self.obstat += obstat
        self._module.set_ob_mean_std(self.obstat.mean, self.obstat.std)

    def optim_step(self, global_g: np.ndarray):
        self.flat_params += self.optim.step(global_g)
