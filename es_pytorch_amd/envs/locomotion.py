"""Synthetic batched locomotion environments (the GPU rollout workload).

The reference trains on PyBullet/MuJoCo locomotion (Hopper, HalfCheetah,
Ant, Humanoid — ``configs/*.json``). Physics engines are not available in
this environment and BASELINE.json prescribes synthetic rollouts of the same
observation/action SHAPES, so these envs implement a smooth, seeded,
nonlinear latent dynamical system with locomotion-shaped semantics:

    s'   = (1-leak)*s + leak * tanh(A s + B a + b0)        (bounded, stable)
    vfwd = s.wv + 0.5 * a.wa                               (forward velocity)
    rew  = vfwd - ctrl * |a|^2 + alive_bonus
    x   += dt * vfwd ;  y += dt * (s.wy)  ;  z = s.wh      (behaviour probe)
    done = z < fall_threshold                              (policy-dependent)

All fixed matrices are drawn from a seeded generator shared by every rank,
so the noiseless phenotype scores identically everywhere. Dynamics are pure
torch elementwise + one (B,S)x(S,S) matmul per step, fixed shapes, no host
sync — hipGraph capture-safe (see ``core/engine.py``).

The dims mirror the MuJoCo/PyBullet originals (Humanoid-v2: 376 obs, 17
actions; Hopper-v3: 11/3; ...), which is what the BASELINE configs name.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd.envs.base import BatchedEnv
from es_pytorch_amd.spaces import Box

# (ob_dim, ac_dim) of the envs the reference configs name
LOCO_SHAPES = {
    "Hopper": (11, 3),
    "Walker2d": (17, 6),
    "HalfCheetah": (17, 6),
    "Ant": (27, 8),
    "Swimmer": (8, 2),
    "Reacher": (11, 2),
    "InvertedPendulum": (4, 1),
    "InvertedDoublePendulum": (11, 1),
    "Humanoid": (376, 17),
    "HumanoidFlagrun": (378, 17),  # Humanoid + 2 goal-relative dims
    "AntFlagrun": (29, 8),         # Ant + 2 goal-relative dims (AntGather-style)
}


class SyntheticLocomotion(BatchedEnv):
    def __init__(self, name: str, batch: int, device="cpu", max_steps: int = 1000,
                 terminate_on_fall: bool = True, env_seed: int = 0xE5, dt: float = 0.05,
                 goal_conditioned: bool = False):
        super().__init__(batch, torch.device(device))
        base = name.replace("Flagrun", "") if goal_conditioned else name
        ob_dim, ac_dim = LOCO_SHAPES[name]
        self.name = name
        self.max_steps = int(max_steps)
        self.terminate_on_fall = terminate_on_fall
        self.goal_conditioned = goal_conditioned
        self.sdim = ob_dim - (2 if goal_conditioned else 0)  # latent state dim
        self.observation_space = Box(-np.inf, np.inf, (ob_dim,))
        self.action_space = Box(-1.0, 1.0, (ac_dim,))

        g = torch.Generator(device="cpu").manual_seed(env_seed + abs(hash(base)) % 100000)
        S, A = self.sdim, ac_dim
        d = self.device
        # the state-transition matrix is held at bf16 precision (stored
        # bf16-rounded fp32 for the torch path, raw bf16 for the HIP kernel)
        # — same compute dtype as the policy weights, and it halves the
        # dominant L2 traffic of the fused rollout step
        self.A = (torch.randn(S, S, generator=g) * (1.1 / np.sqrt(S))) \
            .bfloat16().float().to(d)
        self.A_bf16 = self.A.bfloat16().contiguous()
        self.B = (torch.randn(A, S, generator=g) * (1.0 / np.sqrt(A))).to(d)
        self.b0 = (torch.randn(S, generator=g) * 0.1).to(d)
        self.wv = (torch.randn(S, generator=g) / np.sqrt(S)).to(d)
        self.wa = (torch.randn(A, generator=g) / np.sqrt(A)).to(d)
        self.wy = (torch.randn(S, generator=g) / np.sqrt(S)).to(d)
        self.wh = (torch.randn(S, generator=g) / np.sqrt(S)).to(d)
        self.leak = 0.35
        self.ctrl_cost = 0.05
        self.alive_bonus = 1.0 if terminate_on_fall else 0.0
        self.fall_threshold = -1.6
        self.dt = dt

        self.s = torch.zeros(batch, S, device=d)
        self.pos = torch.zeros(batch, 3, device=d)
        self.goal = torch.zeros(batch, 2, device=d)
        self._steps = torch.zeros(batch, device=d)

    def reset(self, seed: Optional[int] = None) -> torch.Tensor:
        # generate directly on device (avoids a host randn + H2D per generation)
        g = torch.Generator(device=self.device)
        if seed is not None:
            g.manual_seed(int(seed))
        # in-place into persistent buffers: hipGraph-capture-safe
        torch.randn(self.batch, self.sdim, generator=g, device=self.device,
                    out=self.s).mul_(0.1)
        self.pos.zero_()
        self._steps.zero_()
        if self.goal_conditioned:
            ang = torch.rand(self.batch, generator=g, device=self.device) * 2 * np.pi
            r = 15.0
            self.goal.copy_(torch.stack([r * torch.cos(ang), r * torch.sin(ang)], dim=1))
        return self._obs()

    def _obs(self) -> torch.Tensor:
        if self.goal_conditioned:
            rel = self.goal - self.pos[:, :2]
            return torch.cat([self.s, rel * 0.1], dim=1)
        return self.s

    def step(self, actions: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        a = actions.to(self.device).reshape(self.batch, -1).clamp(-1.0, 1.0)
        pre = self.s @ self.A + a @ self.B + self.b0
        self.s.copy_((1 - self.leak) * self.s + self.leak * torch.tanh(pre))

        vfwd = self.s @ self.wv + 0.5 * (a @ self.wa)
        vy = self.s @ self.wy
        h = self.s @ self.wh

        if self.goal_conditioned:
            # reward = progress toward the current goal (flagrun semantics)
            rel = self.goal - self.pos[:, :2]
            dirn = rel / (rel.norm(dim=1, keepdim=True) + 1e-6)
            prog = vfwd * dirn[:, 0] + vy * dirn[:, 1]
            rew = prog - self.ctrl_cost * (a * a).sum(1) + self.alive_bonus
        else:
            rew = vfwd - self.ctrl_cost * (a * a).sum(1) + self.alive_bonus

        self.pos[:, 0] += self.dt * vfwd
        self.pos[:, 1] += self.dt * vy
        self.pos[:, 2] = h
        self._steps += 1

        if self.terminate_on_fall:
            done = (h < self.fall_threshold) | (self._steps >= self.max_steps)
        else:
            done = self._steps >= self.max_steps
        return self._obs(), rew, done

    @property
    def positions(self) -> torch.Tensor:
        return self.pos
