"""PursuitTag co-evolution env + joint runner."""
import numpy as np
import torch

from es_pytorch_amd.envs.multiagent import PursuitTag, make_multiagent
from es_pytorch_amd.rollout import MultiAgentTrainingResult, RewardResult, multi_agent_runner


def test_api_shapes():
    env = make_multiagent("PursuitTag", max_steps=50)
    env.seed(0)
    obs = env.reset()
    assert len(obs) == 2 and obs[0].shape == (8,)
    obs, rews, done, _ = env.step([np.ones(2), -np.ones(2)])
    assert len(rews) == 2
    assert rews[0] == -rews[1]  # strictly competitive (zero-sum shaping)


def test_chase_dynamics():
    env = PursuitTag(max_steps=100)
    env.seed(1)
    env.reset()
    d0 = float(np.linalg.norm(env.p[0] - env.p[1]))
    for _ in range(30):
        toward = env.p[1] - env.p[0]
        toward = toward / (np.linalg.norm(toward) + 1e-9)
        obs, rews, done, _ = env.step([toward, np.zeros(2)])
        if done:
            break
    d1 = float(np.linalg.norm(env.p[0] - env.p[1]))
    assert d1 < d0  # chasing closes distance against a still runner


def test_joint_runner_and_result_split():
    env = PursuitTag(max_steps=20)
    env.seed(2)

    class Pol(torch.nn.Module):
        def forward(self, ob, **kw):
            return torch.zeros(2)

    rews, behv, obs, steps = multi_agent_runner([Pol(), Pol()], env, 20, save_obs=True)
    joint = MultiAgentTrainingResult(rews, behv, obs, steps)
    per = joint.trainingresults(RewardResult)
    assert len(per) == 2
    assert abs(per[0].reward + per[1].reward) < 1e-9  # zero-sum totals


def test_batched_tag_matches_episodic():
    """BatchedPursuitTag reproduces the numpy PursuitTag dynamics."""
    import torch
    from es_pytorch_amd.envs.multiagent import BatchedPursuitTag
    ep = PursuitTag(max_steps=50)
    ep.seed(3)
    ep.reset()
    b = BatchedPursuitTag(1, max_steps=50)
    b.reset(0)
    # force identical initial state
    b.p[0] = torch.from_numpy(ep.p).float()
    b.v[0] = torch.from_numpy(ep.v).float()
    for t in range(10):
        a0 = np.array([0.5, -0.3]) * ((t % 3) - 1)
        a1 = np.array([-0.2, 0.7]) * ((t % 2) * 2 - 1)
        obs_e, rew_e, done_e, _ = ep.step([a0, a1])
        obs_b, rew_b, done_b = b.step([torch.from_numpy(a0).float().reshape(1, 2),
                                       torch.from_numpy(a1).float().reshape(1, 2)])
        np.testing.assert_allclose(obs_b[0][0].numpy(), obs_e[0], atol=1e-5)
        np.testing.assert_allclose(rew_b[0].numpy(), rew_e, atol=1e-5)
        assert bool(done_b[0]) == done_e
        if done_e:
            break
