#!/usr/bin/env python3
"""Replay a saved policy — the equivalent of the reference's
``run_saved.py``: loads a pickled Policy (or a torch.save'd module ``.pt``)
and runs rollouts forever, printing per-episode reward/distance.

  python examples/run_saved.py <env-name> <saved/policy-file> [--episodes N]
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make
from es_pytorch_amd.rollout import run_model


def load_model(path: str) -> torch.nn.Module:
    if path.endswith(".pt"):
        return torch.load(path, weights_only=False)
    return Policy.load(path).pheno()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("env")
    p.add_argument("policy_file")
    p.add_argument("--episodes", type=int, default=0, help="0 = run forever")
    p.add_argument("--max-steps", type=int, default=1000)
    p.add_argument("--render", action="store_true")
    args = p.parse_args()

    env = make(args.env, max_steps=args.max_steps)
    model = load_model(args.policy_file)
    ep = 0
    while args.episodes == 0 or ep < args.episodes:
        rews, behv, obs, steps = run_model(model, env, args.max_steps, None,
                                           render=args.render)
        dist = float(np.linalg.norm(behv[-3:-1]))
        print(f"episode {ep}: reward {sum(rews):.2f} dist {dist:.2f} steps {steps}")
        ep += 1


if __name__ == "__main__":
    main()
