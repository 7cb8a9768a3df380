#!/usr/bin/env python3
"""Determinism / race checker (the framework's race-detection aux tool).

Runs the SAME generation sequence repeatedly from identical seeds and
compares every observable bitwise: flat params, fitnesses, obstat, offsets.
The redundant-update design (reference README.md:10-12) requires identical
inputs -> bitwise-identical updates on every rank; any kernel race or
nondeterministic reduction breaks that silently. This tool makes it loud.

  python tools/race_check.py --repeats 5 --gens 3
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def run_once(gens, graph, fused, pair=None, fp8=None):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    dev = torch.device("cuda", 0)
    torch.manual_seed(77)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 60},
                    "noise": {"tbl_size": 5_000_000, "std": 0.02},
                    "policy": {"layer_sizes": [128, 128], "ac_std": 0.01,
                               "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                               "save_obs_chance": 0.5},
                    "general": {"policies_per_gen": 64, "batch_size": 500, "seed": 4}})
    env = make_batched("Humanoid-v2", 65, dev, max_steps=60, terminate_on_fall=True)
    nn = FeedForward([128, 128], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 5_000_000, len(policy), seed=6, device=dev)
    rs = np.random.RandomState(88)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=graph, fused=fused,
                    pair_rollout=pair, eps_fp8=fp8)
    ranker = CenteredRanker()
    fits_hist = []
    for _ in range(gens):
        tr, ob = eng.step(ranker)
        eng.update_obstat(ob)
        fits_hist.append(np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel().copy())
    return {"flat": policy.flat_params.copy(),
            "fits": np.concatenate(fits_hist),
            "obcount": policy.obstat.count,
            "obsum": policy.obstat.sum.copy(),
            "m": policy.optim.m.copy() if hasattr(policy.optim, "m") else None}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--repeats", type=int, default=5)
    p.add_argument("--gens", type=int, default=3)
    p.add_argument("--no-graph", action="store_true")
    p.add_argument("--pair", action="store_true")
    p.add_argument("--fp8", action="store_true")
    args = p.parse_args()
    assert torch.cuda.is_available(), "race_check needs a GPU"

    base = None
    failures = 0
    for r in range(args.repeats):
        out = run_once(args.gens, graph=not args.no_graph, fused=None,
                       pair=args.pair or None, fp8=args.fp8 or None)
        if base is None:
            base = out
            continue
        for k in ("flat", "fits", "obsum"):
            if not np.array_equal(base[k], out[k]):
                bad = np.flatnonzero(base[k] != out[k])
                print(f"REPEAT {r}: MISMATCH in {k}: {bad.size} elems, "
                      f"first at {bad[:5]}")
                failures += 1
        if base["obcount"] != out["obcount"]:
            print(f"REPEAT {r}: obstat count {base['obcount']} != {out['obcount']}")
            failures += 1
    if failures:
        print(f"RACE CHECK FAILED: {failures} mismatching repeats")
        return 1
    print(f"race check OK: {args.repeats} repeats bitwise-identical "
          f"({args.gens} gens each, graph={not args.no_graph}, "
          f"pair={args.pair}, fp8={args.fp8})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
