"""bench.py contract: JSON output shape, and the torchrun multi-rank flow
(2 CPU ranks over gloo exercising the same allgather/reduce path the driver's
multi-GPU scale run uses)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _parse_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
            "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"]


@pytest.mark.timeout(600)
def test_bench_cpu_single():
    r = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py"), "--cpu",
                        "--steps", "2", "--warmup", "1", "--pop-per-gpu", "4",
                        "--max-steps", "20", "--env", "Hopper-v3", "--layers", "8",
                        "--tbl-size", "200000"],
                       capture_output=True, text=True, timeout=300,
                       env=dict(os.environ, PYTHONPATH=ROOT))
    assert r.returncode == 0, r.stderr[-2000:]
    out = _parse_json_line(r.stdout)
    for k in REQUIRED:
        assert k in out, k
    assert out["value"] > 0 and out["scaling"] == "weak"
    assert out["config"]["global_batch"] == 4
    # encoding transparency: the perturbation format is always reported
    assert out["config"]["eps_encoding"] in ("bf16", "e4m3")


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu():
    """The exact launcher shape the driver uses, world_size=2 on CPU/gloo."""
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--standalone", "--local-addr", "127.0.0.1",
           os.path.join(ROOT, "bench.py"), "--cpu",
           "--gpus", "2", "--steps", "2", "--warmup", "1", "--pop-per-gpu", "4",
           "--max-steps", "20", "--env", "Hopper-v3", "--layers", "8",
           "--tbl-size", "200000"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                       env=dict(os.environ, PYTHONPATH=ROOT))
    assert r.returncode == 0, r.stderr[-3000:]
    out = _parse_json_line(r.stdout)
    assert out["n_gpus"] == 2
    assert out["config"]["global_batch"] == 8  # weak scaling: 4 per rank
    # whole-job env steps: 2 ranks x (8+1 noiseless... counted: 8 members) x 20 x 2 gens
    assert out["value"] > 0
