#!/usr/bin/env python3
"""Generate a checkpoint written by the ACTUAL reference codepath.

Imports the reference repo (/root/reference, read-only) with minimal stand-in
modules for its unavailable dependencies (gym, mpi4py — neither is installed
offline, and neither participates in ``Policy.save``), builds a reference
``Policy`` (``src/core/policy.py:20-28``) around a reference ``FeedForward``
(``src/nn/nn.py:24-41``) and pickles it via the reference's own ``save``
(``src/core/policy.py:43-47``).

The resulting file is vendored at ``tests/fixtures/ref_policy_ckpt`` and is
what ``es_pytorch_amd.core.policy.Policy.load`` must open byte-for-byte —
the cross-load interop contract (BASELINE.json: "same saved-policy
checkpoint format").

Usage: python tools/make_ref_fixture.py [out_path]
"""
import sys
import types

import numpy as np
import torch


def install_stub_modules():
    """gym: only ``gym.Env`` is referenced (type annotations / isinstance);
    mpi4py: ``MPI.Op.Create`` runs at src.nn.obstat import time."""
    gym = types.ModuleType("gym")

    class Env:  # annotation target only
        pass

    gym.Env = Env
    gym.Space = type("Space", (), {})
    sys.modules.setdefault("gym", gym)

    mpi4py = types.ModuleType("mpi4py")
    mpi = types.ModuleType("mpi4py.MPI")

    class _Op:
        @staticmethod
        def Create(fn, commute=False):
            return ("stub-op", fn, commute)

    mpi.Op = _Op
    mpi.Comm = type("Comm", (), {})
    mpi4py.MPI = mpi
    sys.modules.setdefault("mpi4py", mpi4py)
    sys.modules.setdefault("mpi4py.MPI", mpi)


class _EnvShim:
    """Duck-typed env: the reference FeedForward only reads the two spaces'
    shapes (``src/nn/nn.py:33``)."""

    class _Box:
        def __init__(self, shape):
            self.shape = shape

    def __init__(self, ob_dim, ac_dim):
        self.observation_space = self._Box((ob_dim,))
        self.action_space = self._Box((ac_dim,))


def main(out="tests/fixtures/ref_policy_ckpt"):
    install_stub_modules()
    sys.path.insert(0, "/root/reference")
    from src.core.policy import Policy as RefPolicy
    from src.nn.nn import FeedForward as RefFeedForward
    from src.nn.optimizers import Adam as RefAdam

    torch.manual_seed(1234)
    np.random.seed(1234)
    nn = RefFeedForward([8, 8], torch.nn.Tanh(), _EnvShim(4, 2), ac_std=0.01, ob_clip=5)
    policy = RefPolicy(nn, 0.02, RefAdam(len(RefPolicy.get_flat(nn)), 0.01))
    # make the state non-trivial so the round-trip assertions bite
    policy.flat_params += 0.001 * np.arange(len(policy), dtype=np.float32)
    policy.set_nn_params(policy.flat_params)
    policy.obstat.inc(np.arange(4.0), np.arange(4.0) ** 2 + 1.0, 7.0)
    policy.optim.m += 0.25
    policy.optim.v += 0.5
    policy.optim.t = 3

    import os
    folder, fname = os.path.split(out)
    assert fname.startswith("policy-") or True
    policy.save(folder or ".", "ref")  # reference naming: policy-<suffix>
    src = f"{folder or '.'}/policy-ref"
    if src != out:
        os.replace(src, out)
    # sidecar with the expected values the test asserts against
    np.savez(out + ".expected.npz", flat=policy.flat_params,
             obstat_sum=policy.obstat.sum, obstat_sumsq=policy.obstat.sumsq,
             obstat_count=policy.obstat.count, adam_m=policy.optim.m,
             adam_v=policy.optim.v, adam_t=policy.optim.t, std=policy.std)
    print(f"wrote {out} ({len(policy)} params) + expected sidecar")


if __name__ == "__main__":
    main(*sys.argv[1:])
