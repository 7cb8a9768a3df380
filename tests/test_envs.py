"""Environment suite: dynamics sanity, batched/single parity, registry."""
import numpy as np
import torch

from es_pytorch_amd.envs import make, make_batched
from es_pytorch_amd.envs.classic import BatchedCartPole
from es_pytorch_amd.envs.locomotion import LOCO_SHAPES, SyntheticLocomotion


def test_registry_aliases():
    for name in ["CartPole-v1", "HopperBulletEnv-v0", "Hopper-v3", "Humanoid-v2",
                 "HumanoidFlagrunBulletEnv-v0", "AntGatherBulletEnv-v0",
                 "Swimmer-v3", "Reacher-v2", "InvertedDoublePendulum-v2"]:
        env = make(name)
        assert env.observation_space.shape[0] > 0


def test_every_registered_env_steps():
    import torch
    from es_pytorch_amd.envs import make_batched, registry
    for name in registry():
        env = make_batched(name, 3, "cpu")
        ob = env.reset(0)
        assert ob.shape == (3, env.ob_dim), name
        ob2, rew, done = env.step(torch.zeros((3, env.ac_dim)))
        assert ob2.shape == (3, env.ob_dim) and rew.shape == (3,), name


def test_cartpole_api_and_termination():
    env = make("CartPole-v1")
    env.seed(0)
    ob = env.reset()
    assert ob.shape == (4,)
    done, steps = False, 0
    while not done and steps < 600:
        ob, r, done, _ = env.step(np.array([1.0]))  # constant push -> must fall
        assert r == 1.0
        steps += 1
    assert done and steps < 500  # constant force tips the pole well before the limit


def test_cartpole_batched_deterministic():
    b = BatchedCartPole(3)
    ob = b.reset(seed=5)
    b2 = BatchedCartPole(3)
    ob_b2 = b2.reset(seed=5)
    np.testing.assert_allclose(ob.numpy(), ob_b2.numpy())  # seeded reset reproducible
    acts = torch.tensor([[1.0], [1.0], [-1.0]])
    ob2, r, d = b.step(acts)
    assert ob2.shape == (3, 4) and r.shape == (3,) and d.shape == (3,)
    ob2b, _, _ = b2.step(acts)
    np.testing.assert_allclose(ob2.numpy(), ob2b.numpy())


def test_locomotion_shapes_and_determinism():
    for name, (ob_dim, ac_dim) in LOCO_SHAPES.items():
        if name == "HumanoidFlagrun":
            continue
        e = SyntheticLocomotion(name, batch=4, max_steps=50)
        ob = e.reset(seed=1)
        assert ob.shape == (4, ob_dim)
        a = torch.zeros(4, ac_dim)
        ob2, r, d = e.step(a)
        assert ob2.shape == (4, ob_dim) and r.shape == (4,)
        # same seeds -> same rollout
        e2 = SyntheticLocomotion(name, batch=4, max_steps=50)
        e2.reset(seed=1)
        ob2b, r2, _ = e2.step(a)
        np.testing.assert_allclose(ob2.numpy(), ob2b.numpy())
        np.testing.assert_allclose(r.numpy(), r2.numpy())


def test_locomotion_rewards_respond_to_actions():
    e = SyntheticLocomotion("Hopper", batch=64, max_steps=100, terminate_on_fall=False)
    e.reset(seed=0)
    rng = np.random.RandomState(0)
    tot = torch.zeros(64)
    for _ in range(50):
        a = torch.from_numpy(rng.uniform(-1, 1, size=(64, 3)).astype(np.float32))
        _, r, _ = e.step(a)
        tot += r
    # different action sequences must separate fitnesses (ES needs signal)
    assert tot.std().item() > 1e-3


def test_flagrun_goal_conditioned():
    e = make_batched("HumanoidFlagrunBulletEnv-v0", 2, max_steps=20)
    ob = e.reset(seed=3)
    assert ob.shape == (2, 378)
    ob2, r, d = e.step(torch.zeros(2, 17))
    assert ob2.shape == (2, 378)


def test_positions_move():
    e = SyntheticLocomotion("Humanoid", batch=2, max_steps=50, terminate_on_fall=False)
    e.reset(seed=0)
    for _ in range(20):
        e.step(torch.ones(2, 17) * 0.5)
    assert e.positions.abs().sum().item() > 0
