"""Environment registry.

``make(name)`` returns an episodic CPU env; ``make_batched(name, batch,
device)`` returns the population-batched device env used by the GPU engine.
Names accept both MuJoCo-style ids ("Humanoid-v2", "Hopper-v3") and the
PyBullet-style ids the reference configs use ("HopperBulletEnv-v0", ...),
mapping to the built-in synthetic implementations — the reference registers
pybullet envs at import (``src/__init__.py:2-21``); here registration is
explicit and dependency-free.
"""
from __future__ import annotations

import re

from es_pytorch_amd.envs.base import BatchedEnv, Env, SingleFromBatched  # noqa: F401
from es_pytorch_amd.envs.classic import BatchedCartPole, BatchedPendulum
from es_pytorch_amd.envs.locomotion import LOCO_SHAPES, SyntheticLocomotion


def _canonical(name: str) -> str:
    n = re.sub(r"(BulletEnv|PyBulletEnv|MuJoCoEnv)", "", name)
    n = re.sub(r"-v\d+$", "", n)
    return n


def make_batched(name: str, batch: int, device="cpu", **kwargs) -> BatchedEnv:
    c = _canonical(name)
    if c == "CartPole":
        return BatchedCartPole(batch, device)
    if c == "Pendulum":
        return BatchedPendulum(batch, device)
    if c in ("HumanoidFlagrun", "HumanoidFlagrunHarder"):
        kwargs.setdefault("goal_conditioned", True)
        return SyntheticLocomotion("HumanoidFlagrun", batch, device, **kwargs)
    if c in ("AntFlagrun", "AntGather", "AntMaze"):
        # reference flagrun.py's AntGather and ns.json's AntMaze (both
        # hrl_pybullet_envs): goal/target-directed Ant stand-ins
        kwargs.setdefault("goal_conditioned", True)
        return SyntheticLocomotion("AntFlagrun", batch, device, **kwargs)
    if c in LOCO_SHAPES:
        return SyntheticLocomotion(c, batch, device, **kwargs)
    # case-insensitive fallback: PyBullet spells it Walker2D, MuJoCo Walker2d
    for k in LOCO_SHAPES:
        if k.lower() == c.lower():
            return SyntheticLocomotion(k, batch, device, **kwargs)
    raise ValueError(f"unknown env: {name!r} (canonical {c!r}); "
                     f"known: CartPole, Pendulum, {sorted(LOCO_SHAPES)}")


def make(name: str, device="cpu", **kwargs) -> Env:
    return SingleFromBatched(make_batched(name, 1, device, **kwargs))


def registry():
    return ["CartPole-v1", "Pendulum-v1"] + [f"{k}-v3" for k in LOCO_SHAPES]
