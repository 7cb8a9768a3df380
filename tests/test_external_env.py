"""Episodic runner vs an EXTERNAL gym-API environment.

The reference runs on real gym/PyBullet envs (``src/gym/gym_runner.py:33-67``).
Offline, the built-in envs stand in for them — but the episodic runner's
duck-typed compatibility with an env object this repo did NOT define must
still hold. This file vendors a minimal self-contained env speaking the
classic gym protocol and pins down EXACTLY the external surface `run_model`
relies on:

* ``reset() -> ob``  (no seed argument — classic gym)
* ``step(action) -> (ob, reward, done, info)``
* ``seed(s)`` and ``render()`` (optional, only called when used)
* numpy-convertible observations; actions arrive as numpy arrays
* position probe: ``env.unwrapped`` chain + a ``position`` attribute, OR a
  custom ``get_pos_fn`` (the hook replacing the reference's four
  framework-specific probes, ``gym_runner.py:13-30``)
"""
import numpy as np
import torch

from es_pytorch_amd.rollout.runner import env_pos, run_model


class _MinimalBox:
    """Deliberately NOT es_pytorch_amd.spaces.Box — external envs bring
    their own space objects; only ``.shape`` may be assumed."""

    def __init__(self, shape):
        self.shape = shape


class ExternalPointEnv:
    """Self-contained classic-gym-API point-mass: action moves the point,
    reward is -distance-to-goal, episode ends on arrival."""

    def __init__(self, episode_limit=50):
        self.observation_space = _MinimalBox((2,))
        self.action_space = _MinimalBox((2,))
        self.episode_limit = episode_limit
        self._rng = np.random.RandomState(0)
        self._pos = np.zeros(2)
        self._t = 0
        self.render_calls = 0

    # gym protocol ---------------------------------------------------------
    def seed(self, s):
        self._rng = np.random.RandomState(s)
        return [s]

    def reset(self):
        self._pos = np.array([2.0, 0.0])
        self._t = 0
        return self._pos.copy()

    def step(self, action):
        action = np.clip(np.asarray(action, dtype=np.float64), -1, 1)
        self._pos = self._pos + 0.1 * action
        self._t += 1
        dist = float(np.linalg.norm(self._pos))
        done = dist < 0.05 or self._t >= self.episode_limit
        return self._pos.copy(), -dist, done, {}

    def render(self, mode="human"):
        self.render_calls += 1

    # the supported probe surface -----------------------------------------
    @property
    def unwrapped(self):
        return self

    @property
    def position(self):
        return (float(self._pos[0]), float(self._pos[1]), 0.0)


class _ConstModel(torch.nn.Module):
    """Policy that always walks in -x (moves toward the goal)."""

    def forward(self, ob, rs=None):
        return torch.tensor([-1.0, 0.0])


def test_run_model_against_external_env():
    env = ExternalPointEnv()
    env.seed(3)
    rews, behv, obs, steps = run_model(_ConstModel(), env, max_steps=50)

    assert steps < 49, "point walks to goal -> early termination"
    assert len(rews) == steps + 1
    assert obs.shape == (steps + 1, 2)
    # behaviour: 3 floats per step, padded with the final position to
    # 3*max_steps (reference gym_runner.py:66)
    assert len(behv) == 3 * 50
    assert behv[-3:] == behv[3 * steps: 3 * steps + 3]
    # final (x, y) close to the goal
    assert abs(behv[-3]) < 0.1 and abs(behv[-2]) < 0.1


def test_render_hook_called():
    env = ExternalPointEnv(episode_limit=3)
    run_model(_ConstModel(), env, max_steps=3, render=True)
    assert env.render_calls == 3


def test_custom_position_probe():
    """External envs without .position plug in via get_pos_fn — the single
    hook replacing the reference's four framework probes."""

    class NoPosEnv(ExternalPointEnv):
        position = None  # simulate an env without the attribute

    probed = []

    def probe(env):
        probed.append(True)
        return (1.0, 2.0, 3.0)

    env = NoPosEnv(episode_limit=4)
    _, behv, _, _ = run_model(_ConstModel(), env, max_steps=4, get_pos_fn=probe)
    assert probed and behv[:3] == [1.0, 2.0, 3.0]


def test_env_pos_fallback_shapes():
    """env_pos tolerates missing/odd position shapes (external variety)."""
    class P1:
        position = [7.0]
    class NoP:
        pass
    assert env_pos(P1()) == (7.0, 0.0, 0.0)
    assert env_pos(NoP()) == (0.0, 0.0, 0.0)


def test_full_es_generation_on_external_env():
    """End-to-end: es.step trains on the external env (CPU episodic path),
    proving external envs work beyond run_model."""
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core import es
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.run import episodic_fit_fn
    from es_pytorch_amd.utils.rankers import CenteredRanker
    from es_pytorch_amd.utils.reporters import StdoutReporter

    torch.manual_seed(0)
    comm = Comm(torch.device("cpu"))
    cfg = AttrDict({"env": {"name": "external", "max_steps": 10},
                    "noise": {"tbl_size": 100_000, "std": 0.05},
                    "policy": {"layer_sizes": [8], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 4, "batch_size": 100, "seed": 5}})
    env = ExternalPointEnv(episode_limit=10)
    env.seed(5)
    nn = FeedForward([8], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 100_000, len(policy), seed=5)
    rs = np.random.RandomState(5)
    before = policy.flat_params.copy()
    tr, gen_obstat = es.step(cfg, comm, policy, nt, env,
                             episodic_fit_fn(cfg, env, rs), rs,
                             CenteredRanker(), StdoutReporter(comm))
    assert not np.array_equal(before, policy.flat_params), "params updated"
    assert gen_obstat.count > 0
