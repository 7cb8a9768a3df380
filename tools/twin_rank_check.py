#!/usr/bin/env python3
"""Two ranks sharing ONE GPU over gloo: end-to-end validation of the
multi-rank GPU-engine flow without a multi-GPU node.

Exercises exactly the rank-parallel logic the 8-GPU RCCL run uses — per-rank
noise-offset draws, the (fit+, fit-, idx) all-gather, redundant identical
ranking/gradient/Adam on every rank, packed ObStat all-reduce — with real
HIP kernels on cuda:0 in both processes; only the transport differs (gloo
host collectives instead of RCCL, which swaps in transparently when each
rank has its own GPU). Asserts the design invariant: after N generations
every rank holds BITWISE-identical parameters without any parameter
communication (reference es.py:98-101 redundancy contract).

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --standalone --local-addr 127.0.0.1 tools/twin_rank_check.py
"""
import hashlib
import os
import sys

os.environ["LOCAL_RANK"] = "0"  # both ranks deliberately share cuda:0

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    dist.init_process_group(backend="gloo")
    torch.cuda.set_device(0)

    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm, seed_all
    from es_pytorch_amd.utils.rankers import CenteredRanker

    dev = torch.device("cuda", 0)
    comm = Comm(dev)
    assert comm.size == 2, "launch with --nproc-per-node 2"
    rs, _, _ = seed_all(comm, [300, 301])

    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 50},
                    "noise": {"tbl_size": 4_000_000, "std": 0.02},
                    "policy": {"layer_sizes": [64], "ac_std": 0.01, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 0.5},
                    "general": {"policies_per_gen": 32, "batch_size": 500,
                                "seed": 11}})
    B = 2 * (32 // 2 // 2) + 1
    env = make_batched("Humanoid-v2", B, dev, max_steps=50)
    nn = FeedForward([64], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 4_000_000, len(policy), seed=12, device=dev)
    import sys as _sys
    pair = "--pair" in _sys.argv
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True,
                    pair_rollout=True if pair else None)
    ranker = CenteredRanker()

    for _ in range(3):
        eng.step(ranker)
    assert ranker.n_fits_ranked == 32  # both ranks' pairs were gathered

    digest = hashlib.sha256(eng.theta.cpu().numpy().tobytes()).hexdigest()
    opt_digest = hashlib.sha256(eng.m.cpu().numpy().tobytes()).hexdigest()
    all_d = comm.allgather_obj((digest, opt_digest))
    assert all(d == all_d[0] for d in all_d), f"rank divergence: {all_d}"

    # per-rank offsets must DIFFER (distinct rs streams feed distinct pairs)
    offs = comm.allgather_obj(eng.offsets[:eng.pairs].cpu().tolist())
    assert offs[0] != offs[1], "ranks drew identical noise offsets"

    if comm.rank == 0:
        print(f"TWIN-RANK OK (pair={eng.pair_rollout}): 3 gens, "
              f"params sha256={digest[:16]}, moments sha256={opt_digest[:16]}, "
              "identical on both ranks")
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
