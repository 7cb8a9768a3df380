"""Entry scripts end-to-end on CPU with tiny configs (subprocess smoke runs).

The reference never tested its entry scripts (SURVEY.md §4 gap)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, cfg, tmp_path, extra=None, timeout=240):
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    env = dict(os.environ, PYTHONPATH=ROOT)
    r = subprocess.run([sys.executable, os.path.join(ROOT, "examples", script),
                        str(cfg_path)] + (extra or []),
                       capture_output=True, text=True, timeout=timeout,
                       cwd=str(tmp_path), env=env)
    assert r.returncode == 0, f"{script} failed:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"
    return r


def _base(name="CartPole-v1", max_steps=50, gens=2, ppg=4, layers=(8,)):
    return {
        "env": {"name": name, "max_steps": max_steps},
        "noise": {"tbl_size": 100000, "std": 0.05, "std_decay": 0.999, "std_limit": 0.01},
        "policy": {"layer_sizes": list(layers), "ac_std": 0.01, "ac_std_decay": 1.0,
                   "l2coeff": 0.005, "lr": 0.02, "lr_decay": 1.0, "lr_limit": 0.001,
                   "ob_clip": 5, "save_obs_chance": 0.5},
        "general": {"name": "t", "gens": gens, "policies_per_gen": ppg,
                    "batch_size": 100, "seed": 3, "mlflow": False, "n_policies": 2},
        "novelty": {"k": 3, "archive_size": 100, "rollouts": 1},
        "nsr": {"adaptive": True, "progressive": False, "initial_w": 1.0,
                "weight_delta": 0.05, "max_time_since_best": 5,
                "end_progression_gen": 10},
        "experimental": {"elite": 0, "explore_with_large_noise": False,
                         "max_time_since_best": 5},
    }


def test_simple_example(tmp_path):
    _run("simple_example.py", _base(), tmp_path)


def test_obj(tmp_path):
    r = _run("obj.py", _base(), tmp_path)
    assert os.path.exists(tmp_path / "saved")


def test_obj_elite(tmp_path):
    cfg = _base()
    cfg["experimental"]["elite"] = 0.5
    _run("obj.py", cfg, tmp_path)


def test_nsra(tmp_path):
    cfg = _base(name="Hopper-v3", max_steps=30)
    _run("nsra.py", cfg, tmp_path)


def test_nsra_progressive(tmp_path):
    cfg = _base(name="Hopper-v3", max_steps=30)
    cfg["nsr"]["adaptive"] = False
    cfg["nsr"]["progressive"] = True
    _run("nsra.py", cfg, tmp_path)


def test_flagrun(tmp_path):
    cfg = _base(name="HumanoidFlagrunBulletEnv-v0", max_steps=20, layers=(16,))
    _run("flagrun.py", cfg, tmp_path)


def test_multi_agent(tmp_path):
    cfg = _base(name="PursuitTag", max_steps=30)
    _run("multi_agent.py", cfg, tmp_path)


def test_run_saved_roundtrip(tmp_path):
    _run("obj.py", _base(gens=1), tmp_path)
    weights = tmp_path / "saved" / "CartPole-v1-t" / "weights"
    files = [f for f in os.listdir(weights) if f.startswith("policy-")]
    assert files, os.listdir(weights)
    env = dict(os.environ, PYTHONPATH=ROOT)
    r = subprocess.run([sys.executable, os.path.join(ROOT, "examples", "run_saved.py"),
                        "CartPole-v1", str(weights / files[0]), "--episodes", "1",
                        "--max-steps", "20"],
                       capture_output=True, text=True, timeout=120, env=env,
                       cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-1000:]
    assert "episode 0" in r.stdout


def test_batch_run(tmp_path):
    base = _base(gens=1)
    (tmp_path / "base.json").write_text(json.dumps(base))
    batch = {"base_config": "base.json",
             "runs": {"quick-a": {"count": 1,
                                  "overrides": {"policy": {"lr": 0.05}}}}}
    _run("batch_run.py", batch, tmp_path)
    # count decremented to 0
    final = json.loads((tmp_path / "cfg.json").read_text())
    assert final["runs"]["quick-a"]["count"] == 0


def test_viz(tmp_path):
    _run("obj.py", _base(gens=2), tmp_path)
    from es_pytorch_amd.utils import viz
    log = tmp_path / "saved" / "CartPole-v1-t" / "es.log"
    assert log.exists()
    out = viz.graph_log(str(log))
    assert os.path.exists(out)
    fits_dir = tmp_path / "saved" / "CartPole-v1-t" / "fits"
    out2 = viz.graph_fits(str(fits_dir))
    assert os.path.exists(out2)


def test_simple_example_torchrun_world2(tmp_path):
    """simple_example under the torchrun launcher, 2 CPU ranks over gloo."""
    cfg = _base(gens=1)
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--standalone", "--local-addr", "127.0.0.1",
           os.path.join(ROOT, "examples", "simple_example.py"), str(cfg_path)]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=300,
                       cwd=str(tmp_path), env=dict(os.environ, PYTHONPATH=ROOT))
    assert r.returncode == 0, r.stderr[-3000:]
    assert "avg fitness" in r.stdout


def test_obj_resume_from_checkpoint(tmp_path):
    """cfg.policy.load resume path (reference obj.py:39-41)."""
    _run("obj.py", _base(gens=2), tmp_path)
    weights = tmp_path / "saved" / "CartPole-v1-t" / "weights"
    files = sorted(f for f in os.listdir(weights) if f.startswith("policy-"))
    assert files
    cfg = _base(gens=1)
    cfg["policy"]["load"] = str(weights / files[-1])
    cfg["general"]["name"] = "resumed"
    _run("obj.py", cfg, tmp_path)
    assert os.path.exists(tmp_path / "saved" / "CartPole-v1-resumed")


def test_obj_ckpt_ring_resume(tmp_path):
    """general.ckpt_every ring: a relaunch continues from the snapshot."""
    cfg = _base(gens=2)
    cfg["general"]["ckpt_every"] = 1
    _run("obj.py", cfg, tmp_path)
    ring = tmp_path / "saved" / "CartPole-v1-t" / "ckpt"
    assert sorted(os.listdir(ring)) == ["ckpt-1.pkl", "ckpt-2.pkl"]
    cfg["general"]["gens"] = 4
    r = _run("obj.py", cfg, tmp_path)
    assert "resumed from checkpoint at gen 2" in r.stdout
    assert sorted(os.listdir(ring)) == ["ckpt-2.pkl", "ckpt-3.pkl", "ckpt-4.pkl"]


def test_nsra_ckpt_resume(tmp_path):
    """NSRA control state (archive, weights, selection RNG) resumes."""
    cfg = _base(name="Hopper-v3", max_steps=30, gens=2)
    cfg["general"]["ckpt_every"] = 2
    _run("nsra.py", cfg, tmp_path)
    cfg["general"]["gens"] = 3
    r = _run("nsra.py", cfg, tmp_path)
    assert "resumed from checkpoint at gen 2" in r.stdout
