#!/usr/bin/env python3
"""Twin-rank engine integration check: N ranks share ONE GPU.

Launched under torchrun with ES_COMM_BACKEND=gloo (RCCL rejects two ranks on
one device; gloo carries the collectives while ALL compute stays on cuda:0).
Exercises the full distributed generation path end-to-end on hardware:
per-rank offset draws -> HIP pheno/rollout -> fitness-triple all-gather ->
redundant ranking -> gather-GEMV gradient -> fused Adam — then asserts the
reference's core invariant: every rank holds BITWISE identical parameters
after every generation (reference es.py:84-101 replicated-update contract;
the exchange payload is rank-dependent, the merged result must not be).

Exit 0 = all invariants held on every rank.
"""
import hashlib
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.core.engine import GpuEngine
from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.envs import make_batched
from es_pytorch_amd.nn.nn import FeedForward
from es_pytorch_amd.nn.optimizers import Adam
from es_pytorch_amd.parallel.comm import init_comm, seed_all
from es_pytorch_amd.utils.rankers import CenteredRanker


def main():
    assert torch.cuda.is_available(), "twin-rank check needs the GPU"
    comm = init_comm()
    assert comm.size >= 2, "launch under torchrun --nproc-per-node >= 2"
    device = comm.device
    rs, my_seed, global_seed = seed_all(comm, [100 + r for r in range(comm.size)])

    ppg = 64 * comm.size
    cfg = AttrDict({
        "env": {"name": "Humanoid-v2", "max_steps": 40},
        "noise": {"tbl_size": 2_000_000, "std": 0.02},
        "policy": {"layer_sizes": [64, 64], "ac_std": 0.01, "l2coeff": 0.005,
                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
        "general": {"name": "twin", "policies_per_gen": ppg, "batch_size": 500,
                    "seed": global_seed},
    })
    B = 2 * (ppg // comm.size // 2) + 1
    env = make_batched("Humanoid-v2", B, device, max_steps=40,
                       terminate_on_fall=False)
    nn = FeedForward([64, 64], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 2_000_000, len(policy),
                                  seed=global_seed, device=device)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs)
    ranker = CenteredRanker()

    # rank-distinct offset streams, rank-identical initial params
    h0 = comm.allgather_obj(hashlib.sha256(policy.flat_params.tobytes()).hexdigest())
    assert len(set(h0)) == 1, f"initial params differ across ranks: {h0}"

    for gen in range(4):
        eng.step(ranker)
        # the gather really carried every rank's pairs
        assert ranker.n_fits_ranked == ppg, (ranker.n_fits_ranked, ppg)
        # bitwise-identical parameters on every rank after the redundant update
        theta_bytes = eng.theta.cpu().numpy().tobytes()
        hashes = comm.allgather_obj(hashlib.sha256(theta_bytes).hexdigest())
        assert len(set(hashes)) == 1, f"gen {gen}: rank params diverged: {hashes}"
        # distinct noise draws per rank (the exchange is not degenerate)
        offs = comm.allgather_obj(int(eng.offsets[0].item()))
        if gen == 0:
            assert len(set(offs)) == comm.size, f"offset streams collide: {offs}"

    # obstat merge: all ranks agree after dist_inc
    counts = comm.allgather_obj(float(policy.obstat.count))
    assert len(set(counts)) == 1, counts

    # -- phase 2: forced pair rollout + fp8 eps stream (the flagship path)
    eng2 = GpuEngine(cfg, comm, policy, nt, env, rs, pair_rollout=True,
                     eps_fp8=True, use_graph=False)
    assert eng2.pair_rollout and eng2.eps_fp8, "fp8 pair path must engage"
    for gen in range(2):
        eng2.step(ranker)
        hashes = comm.allgather_obj(
            hashlib.sha256(eng2.theta.cpu().numpy().tobytes()).hexdigest())
        assert len(set(hashes)) == 1, f"fp8 gen {gen}: rank params diverged"
    if comm.rank == 0:
        print(f"TWIN-RANK-OK world={comm.size} gens=4+2fp8 ppg={ppg}", flush=True)


if __name__ == "__main__":
    main()
