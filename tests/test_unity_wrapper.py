"""UnityGymWrapper behavior surface, tested against a scripted stand-in
UnityEnvironment (mlagents cannot run offline; the stub replicates the
mlagents_envs API shapes the wrapper touches — behavior specs, decision /
terminal steps, ActionTuple ingestion)."""
import sys
import types

import numpy as np
import pytest
import torch


# ---- stub mlagents_envs -----------------------------------------------------
class _ActionSpec:
    def __init__(self, continuous=0, branches=()):
        self.continuous_size = continuous
        self.discrete_branches = tuple(branches)
        self.discrete_size = len(branches)

    def is_continuous(self):
        return self.continuous_size > 0 and not self.discrete_size

    def is_discrete(self):
        return self.discrete_size > 0 and not self.continuous_size


class _Spec:
    def __init__(self, obs_shapes, action_spec):
        self.observation_shapes = obs_shapes
        self.action_spec = action_spec


class _Steps:
    def __init__(self, obs, reward):
        self.obs = obs          # list per sensor: (n_agents, *shape)
        self.reward = reward

    def __len__(self):
        return len(self.reward)


class FakeUnityEnv:
    """Two teams: 'striker' (2 agents, continuous 3, two sensors) and
    'goalie' (1 agent, discrete [4]). Terminates at terminal_at steps."""

    def __init__(self, terminal_at=None):
        self.behavior_specs = {
            "striker": _Spec([(4,), (2,)], _ActionSpec(continuous=3)),
            "goalie": _Spec([(5,)], _ActionSpec(branches=[4])),
        }
        self.agents = {"striker": 2, "goalie": 1}
        self.terminal_at = terminal_at
        self.t = 0
        self.actions_seen = {}
        self.resets = 0

    def reset(self):
        self.resets += 1
        self.t = 0

    def set_actions(self, team, action_tuple):
        self.actions_seen[team] = np.asarray(action_tuple.continuous)

    def step(self):
        self.t += 1

    def get_steps(self, team):
        n = self.agents[team]
        shapes = self.behavior_specs[team].observation_shapes
        mk = lambda scale: [np.full((n,) + s, scale + i, dtype=np.float32)
                            for i, s in enumerate(shapes)]
        decision = _Steps(mk(float(self.t)), np.full(n, 0.5 * self.t))
        if self.terminal_at is not None and self.t >= self.terminal_at:
            terminal = _Steps(mk(100.0 + self.t), np.full(n, -1.0))
        else:
            terminal = _Steps([np.zeros((0,) + s) for s in shapes], np.zeros(0))
        return decision, terminal


class _ActionTuple:
    def __init__(self, continuous=None, discrete=None):
        self.continuous = continuous
        self.discrete = discrete


@pytest.fixture(autouse=True)
def stub_mlagents(monkeypatch):
    base = types.ModuleType("mlagents_envs.base_env")
    base.ActionTuple = _ActionTuple
    root = types.ModuleType("mlagents_envs")
    root.base_env = base
    monkeypatch.setitem(sys.modules, "mlagents_envs", root)
    monkeypatch.setitem(sys.modules, "mlagents_envs.base_env", base)


def _wrap(**kw):
    from es_pytorch_amd.envs.unity import UnityGymWrapper
    fake = FakeUnityEnv(**kw.pop("fake_kw", {}))
    return UnityGymWrapper(None, 0, env=fake, **kw), fake


def test_tuple_spaces_across_teams():
    from es_pytorch_amd import spaces
    env, fake = _wrap()
    assert env.n_agents == 3
    assert len(env.observation_space) == 3 and len(env.action_space) == 3
    # striker agents: Box(3) actions, obs 4+2=6 (sensor obs are concatenated)
    assert isinstance(env.action_space[0], spaces.Box)
    assert env.action_space[0].shape == (3,)
    assert env.observation_space[0].shape == (6,)
    # goalie: single-branch discrete
    assert isinstance(env.action_space[2], spaces.Discrete)
    assert env.action_space[2].n == 4


def test_step_regroups_and_vstacks_actions():
    env, fake = _wrap()
    obs = env.reset()
    assert len(obs) == 3 and obs[0].shape == (6,)
    a = [np.array([0.1, 0.2, 0.3]), np.array([0.4, 0.5, 0.6]), np.array([2.0])]
    obs, rews, done, info = env.step(a)
    np.testing.assert_allclose(fake.actions_seen["striker"],
                               [[0.1, 0.2, 0.3], [0.4, 0.5, 0.6]])
    np.testing.assert_allclose(fake.actions_seen["goalie"], [[2.0]])
    assert rews.shape == (3,)
    assert not done


def test_terminal_step_supplies_final_obs_and_ends_episode():
    env, fake = _wrap(fake_kw={"terminal_at": 2})
    env.reset()
    a = [np.zeros(3), np.zeros(3), np.zeros(1)]
    _, _, done, _ = env.step(a)
    assert not done
    obs, rews, done, info = env.step(a)
    assert done
    # terminal step's observations (100+t scale) override the decision step
    assert obs[0][0] >= 100.0
    np.testing.assert_allclose(rews, [-1.0] * 3)


def test_max_steps_truncation():
    env, _ = _wrap(max_steps=3)
    env.reset()
    a = [np.zeros(3), np.zeros(3), np.zeros(1)]
    for i in range(3):
        _, _, done, _ = env.step(a)
    assert done  # n >= max_steps (reference unity.py:100)


def test_multi_agent_runner_drives_the_wrapper():
    """The co-evolution rollout loop runs end-to-end against the wrapper."""
    from es_pytorch_amd.rollout.runner import multi_agent_runner

    class _Net(torch.nn.Module):
        def __init__(self, odim):
            super().__init__()
            self.odim = odim

        def forward(self, ob, rs=None):
            return torch.zeros(self.odim)

    env, _ = _wrap(max_steps=5)
    policies = [_Net(3), _Net(3), _Net(1)]
    rews, behv, obs, steps = multi_agent_runner(policies, env, max_steps=5)
    assert steps == 4  # loop index of the final (truncated) step
    assert len(behv) == 3 * 5
