"""Classic-control environments, batched torch implementations.

CartPole follows the standard gym CartPole-v1 dynamics (Euler-integrated
cart-pole, 0.02 s timestep, +-12 deg / +-2.4 m termination, reward 1 per
step, 500-step limit) with a CONTINUOUS Box(-1,1,(1,)) action mapped to the
discrete force by sign — the usual ES adaptation, since the framework's MLP
policies emit continuous actions sized from ``action_space.shape``
(reference ``src/nn/nn.py:33``).

Pendulum-v1 dynamics likewise follow the standard gym formulation.
Both are BatchedEnv (torch, any device) and are auto-adapted to the episodic
API via ``SingleFromBatched``.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd.envs.base import BatchedEnv
from es_pytorch_amd.spaces import Box


class BatchedCartPole(BatchedEnv):
    MAX_STEPS = 500

    def __init__(self, batch: int, device="cpu"):
        super().__init__(batch, torch.device(device))
        self.observation_space = Box(-np.inf, np.inf, (4,))
        self.action_space = Box(-1.0, 1.0, (1,))
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masspole + self.masscart
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.theta_threshold = 12 * 2 * math.pi / 360
        self.x_threshold = 2.4
        self.state = torch.zeros(batch, 4, device=self.device)
        self._steps = torch.zeros(batch, device=self.device)

    def reset(self, seed: Optional[int] = None) -> torch.Tensor:
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(int(seed))
        init = (torch.rand((self.batch, 4), generator=g) * 0.1 - 0.05).to(self.device)
        self.state.copy_(init)
        self._steps.zero_()
        return self.state.clone()

    def step(self, actions: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        a = actions.to(self.device).reshape(self.batch, -1)[:, 0]
        force = torch.where(a > 0, self.force_mag, -self.force_mag)
        x, x_dot, theta, theta_dot = self.state.unbind(1)
        costheta, sintheta = torch.cos(theta), torch.sin(theta)

        temp = (force + self.polemass_length * theta_dot ** 2 * sintheta) / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / \
                   (self.length * (4.0 / 3.0 - self.masspole * costheta ** 2 / self.total_mass))
        xacc = temp - self.polemass_length * thetaacc * costheta / self.total_mass

        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        # in-place into the persistent buffer: hipGraph-capture-safe
        self.state.copy_(torch.stack([x, x_dot, theta, theta_dot], dim=1))
        self._steps += 1

        done = (x.abs() > self.x_threshold) | (theta.abs() > self.theta_threshold) | \
               (self._steps >= self.MAX_STEPS)
        rew = torch.ones(self.batch, device=self.device)
        return self.state.clone(), rew, done

    @property
    def positions(self) -> torch.Tensor:
        p = torch.zeros(self.batch, 3, device=self.device)
        p[:, 0] = self.state[:, 0]
        return p


class BatchedPendulum(BatchedEnv):
    MAX_STEPS = 200

    def __init__(self, batch: int, device="cpu"):
        super().__init__(batch, torch.device(device))
        self.observation_space = Box(-np.inf, np.inf, (3,))
        self.action_space = Box(-2.0, 2.0, (1,))
        self.max_speed = 8.0
        self.max_torque = 2.0
        self.dt = 0.05
        self.g = 10.0
        self.m = 1.0
        self.length = 1.0
        self.th = torch.zeros(batch, device=self.device)
        self.thdot = torch.zeros(batch, device=self.device)
        self._steps = torch.zeros(batch, device=self.device)

    def reset(self, seed: Optional[int] = None) -> torch.Tensor:
        g = torch.Generator(device="cpu")
        if seed is not None:
            g.manual_seed(int(seed))
        self.th.copy_((torch.rand(self.batch, generator=g) * 2 * math.pi - math.pi).to(self.device))
        self.thdot.copy_((torch.rand(self.batch, generator=g) * 2 - 1).to(self.device))
        self._steps.zero_()
        return self._obs()

    def _obs(self) -> torch.Tensor:
        return torch.stack([torch.cos(self.th), torch.sin(self.th), self.thdot], dim=1)

    def step(self, actions: torch.Tensor):
        u = actions.to(self.device).reshape(self.batch, -1)[:, 0].clamp(-self.max_torque,
                                                                        self.max_torque)
        th_norm = torch.atan2(torch.sin(self.th), torch.cos(self.th))
        cost = th_norm ** 2 + 0.1 * self.thdot ** 2 + 0.001 * u ** 2

        newthdot = self.thdot + (3 * self.g / (2 * self.length) * torch.sin(self.th) +
                                 3.0 / (self.m * self.length ** 2) * u) * self.dt
        newthdot = newthdot.clamp(-self.max_speed, self.max_speed)
        self.th.add_(newthdot * self.dt)
        self.thdot.copy_(newthdot)
        self._steps += 1
        done = self._steps >= self.MAX_STEPS
        return self._obs(), -cost, done

    @property
    def positions(self) -> torch.Tensor:
        p = torch.zeros(self.batch, 3, device=self.device)
        p[:, 0] = torch.sin(self.th)
        p[:, 1] = torch.cos(self.th)
        return p
