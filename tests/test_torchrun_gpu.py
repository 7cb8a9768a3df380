"""Multi-rank distributed-path integration on REAL hardware.

The driver's 8-GPU scale run is the only place the RCCL transport executes
with world > 1 (RCCL rejects two ranks on one device). These tests harden
everything AROUND that transport on a single-GPU box: two torchrun ranks
share cuda:0 with gloo carrying the collectives while all compute (HIP
kernels, rollouts, update) runs on the GPU — exactly the code path of an
8-GPU run except for the collective backend.
"""
import json
import os
import socket
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _torchrun(args, nproc=2, timeout=420):
    env = dict(os.environ)
    env.update({"ES_COMM_BACKEND": "gloo", "MASTER_ADDR": "127.0.0.1",
                "HSA_ENABLE_IPC_MODE_LEGACY": env.get("HSA_ENABLE_IPC_MODE_LEGACY", "0")})
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()), *args]
    return subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True,
                          timeout=timeout)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_twin_rank_engine_invariants():
    """2 ranks, 1 GPU: full generation path; bitwise-identical params on
    every rank after every generation (reference es.py:84-101 contract)."""
    r = _torchrun(["tools/twin_rank_engine_check.py"])
    assert r.returncode == 0, f"stdout:\n{r.stdout[-3000:]}\nstderr:\n{r.stderr[-3000:]}"
    assert "TWIN-RANK-OK world=2" in r.stdout


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_bench_two_ranks_one_gpu():
    """bench.py's own multi-rank path (arg plumbing, whole-job aggregation,
    rank-0-only JSON) under torchrun world=2 sharing one GPU."""
    r = _torchrun(["bench.py", "--gpus", "2", "--steps", "3", "--warmup", "1",
                   "--pop-per-gpu", "128", "--max-steps", "50"])
    assert r.returncode == 0, f"stdout:\n{r.stdout[-3000:]}\nstderr:\n{r.stderr[-3000:]}"
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line from rank 0: {r.stdout[-2000:]}"
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["steps"] == 3
    # whole-job env steps: 2 ranks x 128 pop x 50 steps x 3 gens / elapsed
    assert out["value"] > 0
    assert out["config"]["policies_per_gen"] == 256
