"""Unity ML-Agents bridge for multi-agent co-evolution.

Covers the reference's Unity interface (``src/gym/unity.py:14-118``) so the
co-evolution entry (`examples/multi_agent.py`, `multi_agent_runner`) can run
against real ML-Agents builds: per-team behavior specs become a flat TUPLE
of per-agent obs/action spaces, per-agent actions are re-grouped and
vstacked into one ActionTuple per team, and a non-empty terminal step ends
the episode and supplies the final observations/rewards.

``mlagents_envs`` is an optional dependency (Unity cannot run in this
offline environment): the import happens at construction and fails loudly.
The full behavior surface is unit-tested against a scripted stand-in
UnityEnvironment (tests/test_unity_wrapper.py), mirroring how the MLflow
reporter is covered without mlflow.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np

from es_pytorch_amd import spaces


class _Team:
    """Resolved metadata for one Unity behavior (= one team)."""

    def __init__(self, name: str, spec, n_agents: int):
        self.name = name
        self.n_agents = n_agents
        act = spec.action_spec
        if act.is_continuous():
            high = np.ones(act.continuous_size, dtype=np.float32)
            self.agent_action_space = spaces.Box(-high, high)
            self.action_size = act.continuous_size
        elif act.is_discrete():
            branches = list(act.discrete_branches)
            self.action_size = act.discrete_size
            self.agent_action_space = (spaces.Discrete(branches[0])
                                       if act.discrete_size == 1
                                       else spaces.MultiDiscrete(branches))
        else:
            raise ValueError(
                f"behavior {name!r} mixes discrete and continuous actions — "
                "not supported (same restriction as the reference wrapper)")
        # every sensor's observation is flattened and concatenated per agent
        self.obs_size = int(sum(np.prod(s) for s in spec.observation_shapes))
        self.agent_obs_space = spaces.Box(
            -np.inf, np.inf, shape=(self.obs_size,))


class UnityGymWrapper:
    """gym-style facade over a UnityEnvironment (multi-team, multi-agent).

    ``reset()`` returns a list with one observation vector per agent (teams
    flattened in behavior order); ``step(actions)`` takes one action per
    agent in the same order and returns ``(obs_list, rewards, done, info)``.
    ``done`` is True when any team reports a terminal step or after
    ``max_steps`` env steps.
    """

    def __init__(self, name: Optional[str], rank: int, max_steps: int = 2000,
                 render: bool = False, time_scale: float = 50.0, env=None):
        if env is None:
            # the one import that needs the optional dependency
            from mlagents_envs.environment import UnityEnvironment
            from mlagents_envs.side_channel.engine_configuration_channel import \
                EngineConfigurationChannel
            channel = EngineConfigurationChannel()
            channel.set_configuration_parameters(time_scale=time_scale)
            env = UnityEnvironment(name, rank, no_graphics=not render,
                                   side_channels=[channel])
        self._e = env
        self._e.reset()
        self.max_steps = int(max_steps)
        self.n = 0

        self.teams: List[_Team] = []
        for team_name, spec in self._e.behavior_specs.items():
            decision, _ = self._e.get_steps(team_name)
            self.teams.append(_Team(team_name, spec, len(decision.obs[0])))

        self.observation_space = spaces.Tuple(
            [t.agent_obs_space for t in self.teams for _ in range(t.n_agents)])
        self.action_space = spaces.Tuple(
            [t.agent_action_space for t in self.teams for _ in range(t.n_agents)])

    @property
    def n_agents(self) -> int:
        return sum(t.n_agents for t in self.teams)

    def reset(self):
        self._e.reset()
        self.n = 0
        return self._collect()[0]

    def step(self, actions: List[np.ndarray]):
        """One joint step: per-agent actions, regrouped per team."""
        from mlagents_envs.base_env import ActionTuple
        assert len(actions) == self.n_agents, (len(actions), self.n_agents)
        cursor = 0
        for team in self.teams:
            block = np.vstack([np.asarray(a) for a in
                               actions[cursor:cursor + team.n_agents]])
            self._e.set_actions(team.name, ActionTuple(block))
            cursor += team.n_agents
        self._e.step()
        self.n += 1
        return self._collect()

    def _collect(self):
        """(obs_list, rewards, done, info) from each team's current step.

        A non-empty terminal step takes precedence over the decision step
        (its observations/rewards are the episode's final ones — reference
        ``unity.py:102-106``).
        """
        obs: List[np.ndarray] = []
        rews: List[float] = []
        done = self.n >= self.max_steps
        last_step = None
        for team in self.teams:
            decision, terminal = self._e.get_steps(team.name)
            step = terminal if len(terminal) != 0 else decision
            done = done or len(terminal) != 0
            last_step = step
            for a in range(team.n_agents):
                obs.append(np.hstack([np.asarray(o[a]).ravel() for o in step.obs]))
            rews.extend(np.asarray(step.reward).ravel().tolist())
        return obs, np.asarray(rews, dtype=np.float64), done, {"step": last_step}

    def render(self, mode: str = "human"):
        raise RuntimeError("Unity rendering is fixed at construction "
                           "(render=True); it cannot be toggled per step")

    def close(self):
        self._e.close()
