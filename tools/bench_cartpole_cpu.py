#!/usr/bin/env python3
"""BASELINE config 1: CartPole-v1 2-layer MLP objective-ES, world_size=2, CPU.

Spawns 2 gloo ranks on 127.0.0.1 and measures generations/sec of the
episodic ES path (the reference's simple_example.py flow).

  python tools/bench_cartpole_cpu.py --gens 10
"""
import argparse
import json
import multiprocessing as mp
import os
import socket
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def worker(rank, world, port, gens, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank),
                      WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    import torch
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)

    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core import es
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm, seed_all
    from es_pytorch_amd.rollout import RewardResult, run_model
    from es_pytorch_amd.utils.rankers import CenteredRanker
    from es_pytorch_amd.utils.reporters import StdoutReporter

    comm = Comm(torch.device("cpu"))
    cfg = AttrDict({"env": {"name": "CartPole-v1", "max_steps": 500},
                    "general": {"policies_per_gen": 64, "batch_size": 500},
                    "policy": {"l2coeff": 0.005}})
    rs, my_seed, _ = seed_all(comm, [11, 22])
    env = make("CartPole-v1", max_steps=500)
    env.seed(my_seed)
    nn = FeedForward([32, 32], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.02))
    nt = NoiseTable.create_shared(comm, 5_000_000, len(policy), seed=3)
    ranker = CenteredRanker()

    def fit_fn(model, use_noise=True):
        rews, behv, obs, steps = run_model(model, env, 500, rs if use_noise else None)
        return RewardResult(rews, behv, obs, steps)

    reporter = StdoutReporter(comm) if False else _Null()
    es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, reporter)  # warmup
    dist.barrier()
    t0 = time.perf_counter()
    steps = 0
    for _ in range(gens):
        tr, _ = es.step(cfg, comm, policy, nt, env, fit_fn, rs, ranker, reporter)
    dist.barrier()
    dt = time.perf_counter() - t0
    if rank == 0:
        q.put({"metric": "gens/sec, CartPole-v1 MLP(32,32) pop64 world2 CPU",
               "value": round(gens / dt, 3), "unit": "gens/s", "gens": gens,
               "final_noiseless_reward": float(tr.reward)})
    dist.destroy_process_group()


class _Null:
    def print(self, s):
        pass

    def log_gen(self, *a):
        pass


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--gens", type=int, default=10)
    args = ap.parse_args()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    ps = [ctx.Process(target=worker, args=(r, 2, port, args.gens, q)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(600)
    print(json.dumps(q.get()))
