#!/usr/bin/env python3
"""Cross-compile kernel-variant libraries for same-box A/B sweeps.

Builds one _hip_ops .so per (ES_DEPTH_E, ES_DEPTH_T) pair-forward ring
configuration into es_pytorch_amd/ops/variants/ (gitignored; DOES travel
with the gpurun snapshot). On the GPU box:

    ES_HIP_SO=es_pytorch_amd/ops/variants/hip_e8t4.so \
        python tools/kbench.py --pair --graph --steps 1000

Optionally also builds a control .so from a given git revision's kernel
sources (checked out into a temp tree):

    python tools/build_variants.py --revs HEAD~1 --depths 4,4 8,4
"""
import argparse
import os
import subprocess
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from es_pytorch_amd import ops  # noqa: E402

VAR_DIR = os.path.join(os.path.dirname(ops.__file__), "variants")


def build_depth_variant(de: int, dt: int, minwaves: int = 0) -> str:
    os.makedirs(VAR_DIR, exist_ok=True)
    name = f"hip_e{de}t{dt}" + (f"w{minwaves}" if minwaves else "") + ".so"
    out = os.path.join(VAR_DIR, name)
    flags = [f"-DES_DEPTH_E={de}", f"-DES_DEPTH_T={dt}"]
    if minwaves:
        flags.append(f"-DES_PAIR_MINWAVES={minwaves}")
    ops.build_hip(force=True, extra_flags=flags, out=out)
    print("built", out, flags)
    return out


def build_rev_control(rev: str) -> str:
    """Compile the kernel sources as they were at `rev` (control arm)."""
    os.makedirs(VAR_DIR, exist_ok=True)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(VAR_DIR, f"hip_{rev.replace('~', 'p').replace('/', '_')}.so")
    with tempfile.TemporaryDirectory() as td:
        subprocess.run(["git", "archive", rev, "es_pytorch_amd/ops/csrc"],
                       cwd=repo, check=True,
                       stdout=open(os.path.join(td, "a.tar"), "wb"))
        subprocess.run(["tar", "xf", "a.tar"], cwd=td, check=True)
        d = os.path.join(td, "es_pytorch_amd", "ops", "csrc", "hip")
        srcs = sorted(os.path.join(d, f) for f in os.listdir(d) if f.endswith(".hip"))
        hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
        cmd = [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-shared",
               "-fPIC", *srcs, "-o", out]
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(r.stderr[-2000:])
    print("built", out, f"(sources @ {rev})")
    return out


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--depths", nargs="*", default=["4,4", "8,4"],
                    help="DE,DT[,minwaves] combos")
    ap.add_argument("--revs", nargs="*", default=[],
                    help="git revisions to build as control arms")
    a = ap.parse_args()
    for spec in a.depths:
        parts = [int(x) for x in spec.split(",")]
        build_depth_variant(*parts)
    for rev in a.revs:
        build_rev_control(rev)
