"""Regression floor on the pair rollout's update fidelity across sigma.

The pair path's weights are bf16(theta) +- bf16(sigma*eps) instead of
bf16(theta +- sigma*eps) (one extra bf16 rounding); this must stay a
benign perturbation of the UPDATE at every operating sigma, not only the
flagship 0.02. Asserts a cosine floor between
the pair-path and fused-path reconstructed gradients.
"""
import os
import sys

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tools"))


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
@pytest.mark.parametrize("std", [0.005, 0.02, 0.05])
def test_pair_grad_cosine_floor(std):
    """Floor over the sigma band the shipped configs use (0.005-0.05).
    Measured (profiles/pair_fidelity_sweep.txt): cosine 0.95-0.99 in-band
    at 1000 steps; at the out-of-band sigma=0.1 chaos drives pair-vs-fused
    divergence to ~0.88 — documented, not asserted (both paths are equally
    valid approximations there; prefer --no-pair if exact fused numerics
    matter at extreme sigma)."""
    from pair_fidelity import run
    rho, cos = run(horizon=200, std=std, tbl=20_000_000)
    # short-horizon (200-step) floor; tools/pair_fidelity.py tracks the
    # full 1000-step sweep in profiles/
    assert cos > 0.95, f"pair-vs-fused gradient cosine {cos:.4f} at sigma={std}"
    assert rho > 0.9, f"fitness spearman {rho:.4f} at sigma={std}"
