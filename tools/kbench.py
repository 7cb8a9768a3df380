#!/usr/bin/env python3
"""Kernel micro-bench: isolates the fused rollout step kernel for profiling.

Runs the Humanoid flagship shape (pop 1280, MLP 256x256) for a configurable
number of env steps WITHOUT graph capture so rocprofv3 attributes every
dispatch, then prints per-step wall time.

  rocprofv3 --pmc SQ_WAIT_ANY,... -- python tools/kbench.py --steps 30
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--pop", type=int, default=1280)
    p.add_argument("--layers", type=int, nargs="*", default=[256, 256])
    p.add_argument("--env", type=str, default="Humanoid-v2")
    p.add_argument("--graph", action="store_true")
    p.add_argument("--episode", action="store_true", help="whole-episode rollout mode")
    p.add_argument("--chunk", type=int, default=1, help="env steps per launch")
    p.add_argument("--split", action="store_true", help="split-dynamics rollout")
    p.add_argument("--pair", action="store_true", help="antithetic-pair rollout")
    p.add_argument("--dyn-group", type=int, default=5,
                   help="members per dynamics block in split mode")
    p.add_argument("--fp8", action="store_true", help="fp8 sigma*eps stream")
    args = p.parse_args()

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
    comm = Comm(dev)
    torch.manual_seed(0)
    cfg = AttrDict({
        "env": {"name": args.env, "max_steps": args.steps},
        "noise": {"tbl_size": 50_000_000, "std": 0.02},
        "policy": {"layer_sizes": list(args.layers), "ac_std": 0.01, "l2coeff": 0.005,
                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 0.01},
        "general": {"policies_per_gen": args.pop, "batch_size": 500, "seed": 5,
                    "steps_per_launch": args.chunk, "dyn_group": args.dyn_group},
    })
    B = args.pop + 1
    env = make_batched(args.env, B, dev, max_steps=args.steps, terminate_on_fall=False)
    nn = FeedForward(cfg.policy.layer_sizes, torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, cfg.noise.tbl_size, len(policy), seed=5,
                                  device=dev)
    rs = np.random.RandomState(0)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=args.graph,
                    rollout_mode="episode" if args.episode else "step",
                    split_dyn=args.split or None, pair_rollout=args.pair or None,
                    eps_fp8=args.fp8 or None)
    ranker = CenteredRanker()
    eng.step(ranker)  # warmup

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.step(ranker)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"gen: {dt*1e3:.2f} ms  per-step: {dt/args.steps*1e6:.1f} us  "
          f"timings={eng.timings}")


if __name__ == "__main__":
    main()
