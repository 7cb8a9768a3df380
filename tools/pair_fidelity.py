#!/usr/bin/env python3
"""Quantify the pair rollout's numerical fidelity vs the fused path.

The pair path evaluates bf16(theta) +- bf16(sigma*eps) instead of
bf16(theta +- sigma*eps) — a one-ulp-scale weight difference that chaotic
1000-step rollouts amplify into per-member fitness differences. What
matters for ES is the UPDATE: this script measures, at the flagship config,
(a) Spearman rank correlation of the 2*pairs member fitnesses and (b) the
cosine between the reconstructed gradients, pair vs fused, across horizons.
"""
import sys

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])


def run(horizon, std=0.02, tbl=250_000_000):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    out = {}
    for pair in (False, True):
        torch.manual_seed(99)
        comm = Comm(torch.device("cuda", 0))
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": horizon},
                        "noise": {"tbl_size": tbl, "std": std},
                        "policy": {"layer_sizes": [256, 256], "ac_std": 0.01,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 0.01},
                        "general": {"policies_per_gen": 1280, "batch_size": 500,
                                    "seed": 9}})
        env = make_batched("Humanoid-v2", 1281, comm.device, max_steps=horizon,
                           terminate_on_fall=True)
        nn = FeedForward([256, 256], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, std, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, tbl, len(policy), seed=9,
                                      device=comm.device)
        rs = np.random.RandomState(100)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False,
                        pair_rollout=pair)
        ranker = CenteredRanker()
        eng.step(ranker)
        torch.cuda.synchronize()
        fits = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
        out[pair] = (fits, eng.grad.cpu().numpy().copy())
    from scipy.stats import spearmanr
    rho = spearmanr(out[False][0], out[True][0]).correlation
    g0, g1 = out[False][1], out[True][1]
    cos = float(np.dot(g0, g1) / (np.linalg.norm(g0) * np.linalg.norm(g1)))
    print(f"sigma {std:.3f} horizon {horizon:5d}: fitness spearman {rho:.4f}  "
          f"grad cosine {cos:.4f}", flush=True)
    return rho, cos


if __name__ == "__main__":
    # horizon sweep at the flagship sigma + sigma sweep at the flagship
    # horizon (one operating point is not evidence)
    for h in (100, 1000):
        run(h, std=0.02, tbl=50_000_000)
    for s in (0.005, 0.05, 0.1):
        run(1000, std=s, tbl=50_000_000)
