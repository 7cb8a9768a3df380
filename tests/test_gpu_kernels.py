"""HIP kernel numerics vs plain PyTorch fp32 references (run on MI355X).

Each HIP op (noise fill, pheno, fused MLP forward, gather-GEMV, fused Adam)
is compared against an independent fp32 torch implementation of the same op.
The reference repo never needed these (single CPU backend) — SURVEY.md §4.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda", 0)


def _stream(dev):
    return torch.cuda.current_stream(dev).cuda_stream


def test_noise_fill_matches_cpu(dev):
    from es_pytorch_amd.core.noisetable import NoiseTable
    n = 1_000_003
    gpu = NoiseTable.make_noise(n, seed=99, device=dev).cpu()
    cpu = NoiseTable.make_noise(n, seed=99)
    # same Philox stream; libm-vs-ocml ULP differences in the Box-Muller
    # transcendentals only (measured: ~32% of elements differ by ~1 ULP)
    assert torch.allclose(gpu, cpu, atol=2e-5, rtol=1e-5)
    assert (gpu - cpu).abs().max().item() < 2e-5


def test_pheno_kernel(dev):
    from es_pytorch_amd import ops
    P, n = 7, 1000
    theta = torch.randn(n, device=dev)
    table = torch.randn(5000, device=dev)
    offs = torch.tensor([0, 10, 500, 999, 1234, 4000, 0], dtype=torch.int64, device=dev)
    signs = torch.tensor([1, 1, 1, -1, -1, -1, 0], dtype=torch.float32, device=dev)
    out = torch.empty((P, n), dtype=torch.bfloat16, device=dev)
    std = 0.05
    ops.check(ops.hip().es_pheno_bf16(out.data_ptr(), theta.data_ptr(), table.data_ptr(),
                                      offs.data_ptr(), signs.data_ptr(), P, n, n, std,
                                      _stream(dev)), "pheno")
    torch.cuda.synchronize()
    for p in range(P):
        expect = (theta + signs[p] * std * table[offs[p]:offs[p] + n]).bfloat16()
        assert torch.equal(out[p], expect), f"member {p} mismatch"
    # sign-0 slot is exactly bf16(theta)
    assert torch.equal(out[6], theta.bfloat16())


def _torch_mlp_ref(obs, weights_bf16, dims, obmean, obstd, ob_clip):
    """fp32 forward with bf16-rounded weights, per the blob layout (W^T, b)."""
    B = obs.shape[0]
    x = torch.clamp((obs - obmean) / obstd, -ob_clip, ob_clip)
    out = []
    for b in range(B):
        h = x[b]
        off = 0
        for I, O in zip(dims[:-1], dims[1:]):
            Wt = weights_bf16[b, off:off + I * O].float().reshape(I, O)
            off += I * O
            bias = weights_bf16[b, off:off + O].float()
            off += O
            h = torch.tanh(h @ Wt + bias)
        out.append(h)
    return torch.stack(out)


def test_mlp_fwd_kernel(dev):
    from es_pytorch_amd import ops
    torch.manual_seed(0)
    dims = [11, 64, 64, 3]
    n = sum(I * O + O for I, O in zip(dims[:-1], dims[1:]))
    stride = (n + 7) // 8 * 8
    B = 9
    weights = (torch.randn(B, stride, device=dev) * 0.3).bfloat16()
    obs = torch.randn(B, dims[0], device=dev)
    obmean = torch.randn(dims[0], device=dev) * 0.1
    obstd = torch.rand(dims[0], device=dev) + 0.5
    actions = torch.empty(B, dims[-1], device=dev)
    dims_arr = np.array(dims, dtype=np.int32)
    ops.check(ops.hip().es_mlp_fwd(actions.data_ptr(), obs.data_ptr(), weights.data_ptr(),
                                   obmean.data_ptr(), obstd.data_ptr(),
                                   dims_arr.ctypes.data, len(dims), None, 0, B,
                                   5.0, None, stride, 1, B, 0, 1, 0, None, None, _stream(dev)), "mlp_fwd")
    torch.cuda.synchronize()
    ref = _torch_mlp_ref(obs, weights, dims, obmean, obstd, 5.0)
    assert torch.allclose(actions, ref, atol=2e-2, rtol=2e-2), \
        (actions - ref).abs().max().item()


def test_mlp_fwd_odd_output_dim(dev):
    """dims with odd final layer exercise the scalar path."""
    from es_pytorch_amd import ops
    torch.manual_seed(1)
    dims = [376, 256, 256, 17]
    n = sum(I * O + O for I, O in zip(dims[:-1], dims[1:]))
    stride = (n + 7) // 8 * 8
    B = 3
    weights = (torch.randn(B, stride, device=dev) * 0.1).bfloat16()
    obs = torch.randn(B, dims[0], device=dev)
    obmean = torch.zeros(dims[0], device=dev)
    obstd = torch.ones(dims[0], device=dev)
    actions = torch.empty(B, dims[-1], device=dev)
    dims_arr = np.array(dims, dtype=np.int32)
    ops.check(ops.hip().es_mlp_fwd(actions.data_ptr(), obs.data_ptr(), weights.data_ptr(),
                                   obmean.data_ptr(), obstd.data_ptr(),
                                   dims_arr.ctypes.data, len(dims), None, 0, B,
                                   5.0, None, stride, 1, B, 0, 1, 0, None, None, _stream(dev)), "mlp_fwd")
    torch.cuda.synchronize()
    ref = _torch_mlp_ref(obs, weights, dims, obmean, obstd, 5.0)
    assert torch.allclose(actions, ref, atol=2e-2, rtol=2e-2)


def test_mlp_fwd_action_noise_statistics(dev):
    """ac_std noise: mean 0, std ac_std, deterministic in (seed, salt)."""
    from es_pytorch_amd import ops
    dims = [4, 8, 2]
    n = sum(I * O + O for I, O in zip(dims[:-1], dims[1:]))
    stride = (n + 7) // 8 * 8
    B = 4096
    weights = torch.zeros(B, stride, dtype=torch.bfloat16, device=dev)
    obs = torch.zeros(B, 4, device=dev)
    obmean = torch.zeros(4, device=dev)
    obstd = torch.ones(4, device=dev)
    a1 = torch.empty(B, 2, device=dev)
    a2 = torch.empty(B, 2, device=dev)
    dims_arr = np.array(dims, dtype=np.int32)
    seed = torch.tensor([123], dtype=torch.int64, device=dev)
    acstd = torch.tensor([0.5], dtype=torch.float32, device=dev)
    for out, salt in ((a1, 5), (a2, 5)):
        ops.check(ops.hip().es_mlp_fwd(out.data_ptr(), obs.data_ptr(), weights.data_ptr(),
                                       obmean.data_ptr(), obstd.data_ptr(),
                                       dims_arr.ctypes.data, len(dims), seed.data_ptr(),
                                       salt, B, 5.0, acstd.data_ptr(), stride, 1, B,
                                       0, 1, 0, None, None, _stream(dev)), "mlp_fwd")
    torch.cuda.synchronize()
    assert torch.equal(a1, a2)  # same (seed, salt) -> same noise
    noise = a1.flatten()
    assert abs(noise.mean().item()) < 0.02
    assert abs(noise.std().item() - 0.5) < 0.02
    a3 = torch.empty(B, 2, device=dev)
    ops.check(ops.hip().es_mlp_fwd(a3.data_ptr(), obs.data_ptr(), weights.data_ptr(),
                                   obmean.data_ptr(), obstd.data_ptr(),
                                   dims_arr.ctypes.data, len(dims), seed.data_ptr(),
                                   6, B, 5.0, acstd.data_ptr(), stride, 1, B,
                                   0, 1, 0, None, None, _stream(dev)), "mlp_fwd")
    torch.cuda.synchronize()
    assert not torch.equal(a1, a3)  # different salt -> different noise


def test_grad_gather_kernel(dev):
    from es_pytorch_amd import ops
    torch.manual_seed(2)
    table = torch.randn(100_000, device=dev)
    P, n = 333, 4097
    offs = torch.randint(0, 100_000 - n, (P,), dtype=torch.int64, device=dev)
    fits = torch.randn(P, device=dev)
    g = torch.empty(n, device=dev)
    ops.check(ops.hip().es_grad_gather(g.data_ptr(), table.data_ptr(), fits.data_ptr(),
                                       offs.data_ptr(), P, n, 0.0, _stream(dev)), "grad")
    torch.cuda.synchronize()
    rows = torch.stack([table[o:o + n] for o in offs.cpu()])
    expect = fits.cpu() @ rows.cpu()
    assert torch.allclose(g.cpu(), expect, atol=1e-3, rtol=1e-4)


def test_adam_kernel_matches_numpy(dev):
    from es_pytorch_amd import ops
    from es_pytorch_amd.nn.optimizers import Adam
    n = 10_000
    rng = np.random.RandomState(0)
    theta0 = rng.randn(n).astype(np.float32)
    g_np = rng.randn(n).astype(np.float32)
    l2, n_ranked, lr = 0.005, 64.0, 0.01

    theta = torch.from_numpy(theta0.copy()).to(dev)
    m = torch.zeros(n, device=dev)
    v = torch.zeros(n, device=dev)
    g = torch.from_numpy(g_np * n_ranked).to(dev)  # kernel applies gscale
    ref_opt = Adam(n, lr)
    ref_theta = theta0.copy()
    for t in range(1, 4):
        a = lr * np.sqrt(1 - ref_opt.beta2 ** t) / (1 - ref_opt.beta1 ** t)
        ops.check(ops.hip().es_adam_step(theta.data_ptr(), m.data_ptr(), v.data_ptr(),
                                         g.data_ptr(), n, float(a), 0.9, 0.999, 1e-8,
                                         l2, 1.0 / n_ranked, _stream(dev)), "adam")
        # reference semantics: theta += step(l2*theta - grad)
        ref_theta += ref_opt.step(l2 * ref_theta - g_np)
    torch.cuda.synchronize()
    np.testing.assert_allclose(theta.cpu().numpy(), ref_theta, atol=1e-5, rtol=1e-5)


def test_engine_graph_matches_eager(dev):
    """hipGraph-captured rollout == eager rollout for identical seeds."""
    fits = []
    for use_graph in (False, True):
        import numpy as np
        from es_pytorch_amd.config import AttrDict
        from es_pytorch_amd.core.engine import GpuEngine
        from es_pytorch_amd.core.noisetable import NoiseTable
        from es_pytorch_amd.core.policy import Policy
        from es_pytorch_amd.envs import make_batched
        from es_pytorch_amd.nn.nn import FeedForward
        from es_pytorch_amd.nn.optimizers import Adam
        from es_pytorch_amd.parallel.comm import Comm
        from es_pytorch_amd.utils.rankers import CenteredRanker

        torch.manual_seed(3)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 40},
                        "noise": {"tbl_size": 1_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64, 64], "ac_std": 0.0, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 1}})
        env = make_batched("Hopper-v3", 9, dev, max_steps=40, terminate_on_fall=False)
        nn = FeedForward([64, 64], torch.nn.Tanh(), env, 0.0, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=5, device=dev)
        rs = np.random.RandomState(11)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=use_graph)
        ranker = CenteredRanker()
        eng.step(ranker)
        fits.append(np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel())
    np.testing.assert_allclose(fits[0], fits[1], rtol=1e-5, atol=1e-5)


def test_engine_smoke_and_param_motion(dev):
    import __graft_entry__
    __graft_entry__.smoke()


def test_fused_loco_matches_torch_path(dev):
    """rollout_loco.hip (fused forward+dynamics+bookkeeping) vs the generic
    torch env path: same seeds -> same fitnesses to fp32 tolerance."""
    results = {}
    for fused in (False, True):
        import numpy as np
        from es_pytorch_amd.config import AttrDict
        from es_pytorch_amd.core.engine import GpuEngine
        from es_pytorch_amd.core.noisetable import NoiseTable
        from es_pytorch_amd.core.policy import Policy
        from es_pytorch_amd.envs import make_batched
        from es_pytorch_amd.nn.nn import FeedForward
        from es_pytorch_amd.nn.optimizers import Adam
        from es_pytorch_amd.parallel.comm import Comm
        from es_pytorch_amd.utils.rankers import CenteredRanker

        torch.manual_seed(4)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 30},
                        "noise": {"tbl_size": 1_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64, 64], "ac_std": 0.01,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 1}})
        env = make_batched("Humanoid-v2", 9, dev, max_steps=30, terminate_on_fall=False)
        nn = FeedForward([64, 64], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=5, device=dev)
        rs = np.random.RandomState(21)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False, fused=fused)
        ranker = CenteredRanker()
        tr, obstat = eng.step(ranker)
        results[fused] = {
            "fits": np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel(),
            "ob_mean": obstat.mean.copy(),
            "count": obstat.count,
            "flat": policy.flat_params.copy(),
        }
    a, b = results[False], results[True]
    np.testing.assert_allclose(a["fits"], b["fits"], rtol=1e-3, atol=1e-2)
    assert a["count"] == b["count"]
    np.testing.assert_allclose(a["ob_mean"], b["ob_mean"], rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(a["flat"], b["flat"], rtol=1e-3, atol=1e-4)


def test_fused_loco_termination(dev):
    """Fall termination freezes reward/steps/behaviour like the torch path."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    outs = {}
    for fused in (False, True):
        torch.manual_seed(4)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 60},
                        "noise": {"tbl_size": 500_000, "std": 0.5},
                        "policy": {"layer_sizes": [64], "ac_std": 0.0,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 16, "batch_size": 500, "seed": 1}})
        # fall_threshold raised so terminations actually trigger
        env = make_batched("Humanoid-v2", 17, dev, max_steps=60, terminate_on_fall=True)
        env.fall_threshold = -0.05
        nn = FeedForward([64], torch.nn.Tanh(), env, 0.0, 5)
        policy = Policy(nn, 0.5, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 500_000, len(policy), seed=6, device=dev)
        rs = np.random.RandomState(3)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False, fused=fused)
        eng.step(CenteredRanker())
        outs[fused] = (eng.member_steps.cpu().numpy().copy(),
                       eng.rew_total.cpu().numpy().copy(),
                       eng.behv.cpu().numpy().copy())
    steps_a, rew_a, behv_a = outs[False]
    steps_b, rew_b, behv_b = outs[True]
    assert steps_a.min() < 60, "no member terminated; test is vacuous"
    # fp reassociation between the two implementations moves h values across
    # the fall threshold for some members (and the torch path's hipBLASLt
    # GEMM picks varying algorithms run-to-run, so the exact mismatch set is
    # not even stable); require a clear majority to terminate identically —
    # the bookkeeping semantics, not bitwise dynamics, are under test
    same = steps_a == steps_b
    assert same.mean() > 0.6, (steps_a, steps_b)
    # the recurrent dynamics amplify 1-ulp differences exponentially over the
    # horizon (and the torch arm's hipBLASLt picks varying algorithms
    # run-to-run), so rewards only match statistically; an occasional
    # outlier member is physics, not a bug — the EXACT bookkeeping contract
    # lives in tests/test_termination_exact.py (same-kernel arms, bitwise)
    close = np.isclose(rew_a[same], rew_b[same], rtol=5e-2, atol=5e-2)
    assert close.mean() > 0.8, (rew_a[same], rew_b[same])
    assert np.abs(rew_a[same] - rew_b[same]).max() < 10.0


def test_engine_checkpoint_resume(dev, tmp_path):
    """Engine -> reference-format pickle -> fresh engine: training state
    (params, Adam moments, obstat) survives the round trip."""
    import os

    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(9)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 20},
                    "noise": {"tbl_size": 500_000, "std": 0.02},
                    "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 1}})
    env = make_batched("Hopper-v3", 9, dev, max_steps=20, terminate_on_fall=False)
    nn = FeedForward([32], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 500_000, len(policy), seed=2, device=dev)
    rs = np.random.RandomState(7)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False)
    ranker = CenteredRanker()
    for _ in range(2):
        tr, obstat = eng.step(ranker)
        eng.update_obstat(obstat)
    eng.sync_host()  # full sync incl. optimizer moments
    policy.save(str(tmp_path), "ckpt")

    loaded = Policy.load(os.path.join(str(tmp_path), "policy-ckpt"))
    np.testing.assert_array_equal(loaded.flat_params, policy.flat_params)
    np.testing.assert_array_equal(loaded.optim.m, policy.optim.m)
    assert loaded.optim.t == policy.optim.t == 2
    assert loaded.obstat.count == policy.obstat.count

    eng2 = GpuEngine(cfg, comm, loaded, nt, env, rs, use_graph=False)
    # device state reconstructed from the checkpoint
    np.testing.assert_allclose(eng2.theta.cpu().numpy(), eng.theta.cpu().numpy())
    np.testing.assert_allclose(eng2.m.cpu().numpy(), eng.m.cpu().numpy(), atol=1e-7)
    eng2.step(ranker)  # resumes without error
    assert np.isfinite(loaded.flat_params).all()


def test_binned_policy_engine(dev):
    """K9 binned-action decode: FFBinned on the engine, fused vs torch-path
    parity and agreement with the episodic FFBinned forward."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FFBinned
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    fits = {}
    for fused in (False, True):
        torch.manual_seed(6)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 25},
                        "noise": {"tbl_size": 500_000, "std": 0.05},
                        "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 1}})
        env = make_batched("Hopper-v3", 9, dev, max_steps=25, terminate_on_fall=False)
        nn = FFBinned([32], torch.nn.Tanh(), env, n_bins=5, ob_clip=5)
        policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 500_000, len(policy), seed=4, device=dev)
        rs = np.random.RandomState(13)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False, fused=fused)
        assert eng.bins == 5
        ranker = CenteredRanker()
        eng.step(ranker)
        fits[fused] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2)


def test_engine_eps_per_policy(dev):
    """eps_per_policy > 1: episode-averaged member fitness (obj.py:56-63)."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(8)
    comm = Comm(dev)
    eps = 3
    cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 25},
                    "noise": {"tbl_size": 500_000, "std": 0.05},
                    "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 1,
                                "eps_per_policy": eps}})
    M = 9
    env = make_batched("Hopper-v3", M * eps, dev, max_steps=25, terminate_on_fall=False)
    nn = FeedForward([32], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 500_000, len(policy), seed=4, device=dev)
    rs = np.random.RandomState(17)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False)
    assert eng.eps == eps and eng.B == M * eps and eng.M == M
    ranker = CenteredRanker()
    eng.step(ranker)
    # episode-averaged fitness equals the mean of the member's slot rewards
    rt = eng.rew_total.view(M, eps).cpu().numpy()
    got = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    np.testing.assert_allclose(got, rt[:M - 1].mean(1), rtol=1e-5)
    # slot rewards within a member differ (different env inits) -> real averaging
    assert np.abs(rt[:, 0] - rt[:, 1]).max() > 1e-4


def test_engine_generic_env_graph(dev):
    """Generic (non-locomotion) torch-env engine path under hipGraph capture:
    CartPole batched env + standalone forward kernel, graph == eager."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    fits = {}
    for use_graph in (False, True):
        torch.manual_seed(12)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "CartPole-v1", "max_steps": 40},
                        "noise": {"tbl_size": 300_000, "std": 0.05},
                        "policy": {"layer_sizes": [16], "ac_std": 0.0, "l2coeff": 0.005,
                                   "lr": 0.02, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 100, "seed": 2}})
        env = make_batched("CartPole-v1", 9, dev)
        nn = FeedForward([16], torch.nn.Tanh(), env, 0.0, 5)
        policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.02))
        nt = NoiseTable.create_shared(comm, 300_000, len(policy), seed=9, device=dev)
        rs = np.random.RandomState(31)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=use_graph)
        assert not eng.fused  # CartPole takes the generic torch-env path
        ranker = CenteredRanker()
        eng.step(ranker)
        fits[use_graph] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-5, atol=1e-5)
    assert np.abs(fits[True]).sum() > 0  # episodes produced reward


def test_integ_gauss_engine(dev):
    """FFIntegGausAction on the engine (act_mode 2): the net's own first
    output is the action std; fused and generic kernel paths agree."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FFIntegGausAction
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.spaces import Box
    from es_pytorch_amd.utils.rankers import CenteredRanker

    fits = {}
    for fused in (False, True):
        torch.manual_seed(14)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 25},
                        "noise": {"tbl_size": 400_000, "std": 0.05},
                        "policy": {"layer_sizes": [32], "ac_std": 0.0, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 100, "seed": 2}})
        env = make_batched("Hopper-v3", 9, dev, max_steps=25, terminate_on_fall=False)

        # unified output contract: the net sizes its own output adim+1
        nn = FFIntegGausAction([32], torch.nn.Tanh(), env, ac_std=0.0, ob_clip=5)
        policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 400_000, len(policy), seed=8, device=dev)
        rs = np.random.RandomState(23)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False, fused=fused)
        assert eng.act_mode == 2
        ranker = CenteredRanker()
        eng.step(ranker)
        fits[fused] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2)


def _engine_pair(dev, env_name, layers, pop, max_steps, seed=20, B=None):
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    fits = {}
    for fused in (False, True):
        torch.manual_seed(seed)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": env_name, "max_steps": max_steps},
                        "noise": {"tbl_size": 3_000_000, "std": 0.02},
                        "policy": {"layer_sizes": list(layers), "ac_std": 0.01,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": pop, "batch_size": 500,
                                    "seed": 1}})
        env = make_batched(env_name, pop + 1, dev, max_steps=max_steps,
                           terminate_on_fall=False)
        nn = FeedForward(list(layers), torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 3_000_000, len(policy), seed=seed + 1,
                                      device=dev)
        rs = np.random.RandomState(seed + 2)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False, fused=fused)
        ranker = CenteredRanker()
        eng.step(ranker)
        fits[fused] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    return fits


def test_wide_hidden_layer_parity(dev):
    """1024-wide hidden layer: OCT=128/PART=2 vector-path shape."""
    import numpy as np
    fits = _engine_pair(dev, "Humanoid-v2", [1024], pop=4, max_steps=15)
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2)


def test_odd_state_dim_env_parity(dev):
    """Walker2d (S=17): scalar bf16 dynamics path (S % 8 != 0)."""
    import numpy as np
    fits = _engine_pair(dev, "Walker2d-v3", [64, 64], pop=8, max_steps=20, seed=30)
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2)


def test_minimal_population(dev):
    """pairs=1 (B=3): smallest legal engine shape incl. side-stream episode."""
    import numpy as np
    fits = _engine_pair(dev, "Hopper-v3", [16], pop=2, max_steps=10, seed=40)
    assert fits[True].shape == (2,)
    np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2)


def test_engine_bitwise_determinism(dev):
    """Same seeds + same config -> bitwise-identical parameters after 3 gens.

    This is the multi-rank redundant-update contract (reference
    utils.py:69-70 semantics): ranks rely on identical inputs producing
    identical updates; any nondeterministic kernel would break it."""
    import numpy as np
    flats = []
    for _ in range(2):
        from es_pytorch_amd.config import AttrDict
        from es_pytorch_amd.core.engine import GpuEngine
        from es_pytorch_amd.core.noisetable import NoiseTable
        from es_pytorch_amd.core.policy import Policy
        from es_pytorch_amd.envs import make_batched
        from es_pytorch_amd.nn.nn import FeedForward
        from es_pytorch_amd.nn.optimizers import Adam
        from es_pytorch_amd.parallel.comm import Comm
        from es_pytorch_amd.utils.rankers import CenteredRanker

        torch.manual_seed(50)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 30},
                        "noise": {"tbl_size": 2_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64, 64], "ac_std": 0.01,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 0.5},
                        "general": {"policies_per_gen": 16, "batch_size": 500,
                                    "seed": 3}})
        env = make_batched("Humanoid-v2", 17, dev, max_steps=30,
                           terminate_on_fall=False)
        nn = FeedForward([64, 64], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 2_000_000, len(policy), seed=7, device=dev)
        rs = np.random.RandomState(51)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True)
        ranker = CenteredRanker()
        for _ in range(3):
            tr, ob = eng.step(ranker)
            eng.update_obstat(ob)
        flats.append(policy.flat_params.copy())
    np.testing.assert_array_equal(flats[0], flats[1])


def test_multi_agent_gpu_engine(dev):
    """GPU-batched co-evolution: zero-sum rewards, per-policy updates move."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.ma_engine import MultiAgentGpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs.multiagent import BatchedPursuitTag
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(60)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "PursuitTag", "max_steps": 40},
                    "noise": {"tbl_size": 500_000, "std": 0.05},
                    "policy": {"layer_sizes": [16], "ac_std": 0.01, "l2coeff": 0.005,
                               "lr": 0.02, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 16, "batch_size": 100, "seed": 6}})
    env = BatchedPursuitTag(17, dev, max_steps=40)

    class _View:
        def __init__(self, i):
            self.observation_space = env.observation_space[i]
            self.action_space = env.action_space[i]

    policies = []
    for i in range(2):
        nn = FeedForward([16], torch.nn.Tanh(), _View(i), 0.01, 5)
        policies.append(Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.02)))
    nt = NoiseTable.create_shared(comm, 500_000, len(policies[0]), seed=11, device=dev)
    rs = np.random.RandomState(61)
    eng = MultiAgentGpuEngine(cfg, comm, policies, nt, env, rs)
    flats0 = [p.flat_params.copy() for p in policies]
    for _ in range(2):
        rankers = [CenteredRanker(), CenteredRanker()]
        noiseless, obstats = eng.step(rankers)
        eng.update_obstats(obstats)
    # strictly competitive env: per-instance rewards sum to ~0
    total = eng.rew_total.sum(dim=1)
    assert total.abs().max().item() < 1e-2
    for p, f0 in zip(policies, flats0):
        assert not np.array_equal(p.flat_params, f0)
        assert np.isfinite(p.flat_params).all()
    assert eng.timings["env_steps"] > 0


def test_episode_rollout_mode_parity(dev):
    """rollout_mode='episode' (whole episodes per launch) == per-step mode."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    fits = {}
    for mode in ("step", "episode"):
        torch.manual_seed(70)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 30},
                        "noise": {"tbl_size": 1_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64], "ac_std": 0.01, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 2}})
        env = make_batched("Humanoid-v2", 9, dev, max_steps=30, terminate_on_fall=False)
        nn = FeedForward([64], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=3, device=dev)
        rs = np.random.RandomState(71)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True,
                        rollout_mode=mode)
        ranker = CenteredRanker()
        eng.step(ranker)
        fits[mode] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    np.testing.assert_allclose(fits["step"], fits["episode"], rtol=1e-5, atol=1e-5)


def test_device_rank_matches_host_rank(dev):
    """K4 device argsort ranking == host numpy CenteredRanker (tie-free fits)."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    class _HostCentered(CenteredRanker):
        pass  # subclass -> engine's `type(...) is CenteredRanker` gate is False

    thetas = {}
    for cls in (CenteredRanker, _HostCentered):
        torch.manual_seed(80)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 25},
                        "noise": {"tbl_size": 1_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64], "ac_std": 0.01, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 0.5},
                        "general": {"policies_per_gen": 16, "batch_size": 500,
                                    "seed": 4}})
        env = make_batched("Humanoid-v2", 17, dev, max_steps=25,
                           terminate_on_fall=False)
        nn = FeedForward([64], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=12, device=dev)
        rs = np.random.RandomState(81)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False)
        ranker = cls()
        for _ in range(2):
            eng.step(ranker)
        thetas[cls.__name__] = policy.flat_params.copy()
        assert eng is not None
    # tolerance note: fitness TIES rank differently between numpy quicksort
    # (host) and torch stable sort (device) — both self-consistent across
    # ranks; a tied pair swaps two centered ranks, perturbing theta by
    # ~lr/pop on a handful of elements
    np.testing.assert_allclose(thetas["CenteredRanker"], thetas["_HostCentered"],
                               rtol=5e-3, atol=1e-4)


def test_sgd_kernel_matches_numpy(dev):
    from es_pytorch_amd import ops
    from es_pytorch_amd.nn.optimizers import SGD
    n = 10_000
    rng = np.random.RandomState(1)
    theta0 = rng.randn(n).astype(np.float32)
    g_np = rng.randn(n).astype(np.float32)
    l2, n_ranked, lr, mom = 0.005, 32.0, 0.02, 0.9

    theta = torch.from_numpy(theta0.copy()).to(dev)
    v = torch.zeros(n, device=dev)
    g = torch.from_numpy(g_np * n_ranked).to(dev)  # kernel applies gscale
    ref_opt = SGD(n, lr, momentum=mom)
    ref_theta = theta0.copy()
    for _ in range(3):
        ops.check(ops.hip().es_sgd_step(theta.data_ptr(), v.data_ptr(), g.data_ptr(),
                                        n, lr, mom, l2, 1.0 / n_ranked, _stream(dev)),
                  "sgd")
        # reference semantics: theta += step(l2*theta - grad)
        ref_theta += ref_opt.step(l2 * ref_theta - g_np)
    torch.cuda.synchronize()
    np.testing.assert_allclose(theta.cpu().numpy(), ref_theta, atol=1e-5, rtol=1e-5)


def test_engine_sgd_optimizer(dev):
    """Engine with SGD optimizer end-to-end."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import SGD
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(90)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 15},
                    "noise": {"tbl_size": 300_000, "std": 0.05},
                    "policy": {"layer_sizes": [16], "ac_std": 0.0, "l2coeff": 0.005,
                               "lr": 0.02, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": 8, "batch_size": 100, "seed": 1}})
    env = make_batched("Hopper-v3", 9, dev, max_steps=15, terminate_on_fall=False)
    nn = FeedForward([16], torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, 0.05, SGD(len(Policy.get_flat(nn)), 0.02))
    nt = NoiseTable.create_shared(comm, 300_000, len(policy), seed=13, device=dev)
    rs = np.random.RandomState(91)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False)
    flat0 = policy.flat_params.copy()
    eng.step(CenteredRanker())
    assert not np.array_equal(policy.flat_params, flat0)
    assert np.isfinite(policy.flat_params).all()


def test_engine_all_objectives(dev):
    """Every engine objective produces finite, sensible fitnesses."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker, MultiObjectiveRanker

    for objective in ("reward", "mean_reward", "dist", "xdist", "ns", "nsr"):
        torch.manual_seed(95)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 15},
                        "noise": {"tbl_size": 300_000, "std": 0.05},
                        "policy": {"layer_sizes": [16], "ac_std": 0.0, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 100,
                                    "seed": 1}})
        env = make_batched("Hopper-v3", 9, dev, max_steps=15, terminate_on_fall=False)
        nn = FeedForward([16], torch.nn.Tanh(), env, 0.0, 5)
        policy = Policy(nn, 0.05, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 300_000, len(policy), seed=14, device=dev)
        rs = np.random.RandomState(96)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, objective=objective,
                        use_graph=False)
        if objective in ("ns", "nsr"):
            eng.archive = torch.randn(8, 2, dtype=torch.float64, device=dev)
        ranker = MultiObjectiveRanker(CenteredRanker(), 0.5) if objective == "nsr" \
            else CenteredRanker()
        eng.step(ranker)
        fits = np.concatenate([ranker.fits_pos, ranker.fits_neg])
        assert np.isfinite(fits).all(), objective
        if objective == "dist":
            assert (fits >= 0).all()  # distances are non-negative
        if objective == "ns":
            assert (fits >= 0).all()  # novelty is a mean distance
        if objective == "nsr":
            assert fits.shape[1] == 2


def test_split_dyn_bitwise_matches_fused(dev):
    """Split-dynamics rollout (forward kernel + shared-A dynamics kernel,
    es_loco_step_split) produces BITWISE-identical trajectories and updates
    to the fused single-kernel step: same per-member (oi, ip) tiling, same
    depth-4 pipeline, same loco_dyn_finish epilogue."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    out = {}
    # dyn_group=5 with 8 perturbed members -> a 3-member tail block
    for split in (False, True):
        torch.manual_seed(86)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 25},
                        "noise": {"tbl_size": 1_000_000, "std": 0.02},
                        "policy": {"layer_sizes": [64], "ac_std": 0.01, "l2coeff": 0.005,
                                   "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500, "seed": 2,
                                    "dyn_group": 5}})
        env = make_batched("Humanoid-v2", 9, dev, max_steps=25, terminate_on_fall=True)
        nn = FeedForward([64], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=3, device=dev)
        rs = np.random.RandomState(87)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True, split_dyn=split)
        assert eng.split_dyn == split
        ranker = CenteredRanker()
        for _ in range(2):
            eng.step(ranker)
        torch.cuda.synchronize(dev)
        out[split] = (eng.theta.cpu().clone(), eng.rew_total.cpu().clone(),
                      np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel())
    assert torch.equal(out[False][0], out[True][0])  # params bitwise equal
    assert torch.equal(out[False][1], out[True][1])  # rewards bitwise equal
    np.testing.assert_array_equal(out[False][2], out[True][2])


def test_new_env_shapes_parity(dev):
    """Swimmer (S=8: OCT=1 octet dynamics tiling), InvertedPendulum (S=4:
    scalar fallback), AntFlagrun (goal-conditioned, S=27): fused == torch."""
    import numpy as np
    for env_name, layers in (("Swimmer-v3", [32]), ("InvertedPendulum-v2", [16]),
                             ("AntFlagrun-v3", [32])):
        fits = _engine_pair(dev, env_name, layers, pop=6, max_steps=15, seed=61)
        np.testing.assert_allclose(fits[False], fits[True], rtol=1e-3, atol=1e-2,
                                   err_msg=env_name)


def test_engine_exact_checkpoint_resume(dev, tmp_path):
    """RunCheckpointer on the GPU engine: 4 gens straight == 2 gens + save +
    restore-into-fresh-engine + 2 gens, bitwise (params, moments, rewards).
    Covers the prefetched-offsets hazard: the side-stream offset prefetch
    consumes rs draws at the end of a step, and the snapshot must carry the
    drawn values rather than let the resumed run re-draw."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.checkpoint import RunCheckpointer
    from es_pytorch_amd.utils.rankers import CenteredRanker

    def fresh():
        torch.manual_seed(50)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 20},
                        "noise": {"tbl_size": 400_000, "std": 0.02},
                        "policy": {"layer_sizes": [32], "ac_std": 0.01,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 0.5},
                        "general": {"policies_per_gen": 8, "batch_size": 500,
                                    "seed": 4}})
        env = make_batched("Hopper-v3", 9, dev, max_steps=20)
        nn = FeedForward([32], torch.nn.Tanh(), env, 0.01, 5)
        policy = Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 400_000, len(policy), seed=5, device=dev)
        rs = np.random.RandomState(51)
        eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True)
        return comm, cfg, env, rs, policy, eng

    def run(objs, n):
        *_, eng = objs
        ranker = CenteredRanker()
        obstats = []
        for _ in range(n):
            tr, ob = eng.step(ranker)
            eng.update_obstat(ob)
            obstats.append(tr.reward)
        return obstats

    a = fresh()
    rews_a = run(a, 4)
    a[5].sync_host(light=False)

    b = fresh()
    run(b, 2)
    ck = RunCheckpointer(str(tmp_path / "ring"), b[0])
    ck.save(2, b[4], b[3], cfg=b[1], engine=b[5])

    c = fresh()
    comm_c, cfg_c, env_c, rs_c, policy_c, eng_c = c
    ck2 = RunCheckpointer(str(tmp_path / "ring"), comm_c)
    state = ck2.load()
    next_gen, _ = ck2.restore(state, policy_c, rs_c, cfg=cfg_c, engine=eng_c)
    assert next_gen == 2 and eng_c.gen == 2
    rews_c = run(c, 2)
    eng_c.sync_host(light=False)

    np.testing.assert_array_equal(a[4].flat_params, policy_c.flat_params)
    np.testing.assert_array_equal(a[4].optim.m, policy_c.optim.m)
    assert a[4].optim.t == policy_c.optim.t
    assert rews_a[2:] == rews_c  # resumed generations bitwise-reproduce


def test_twin_rank_engine_identity(dev):
    """Two ranks sharing ONE GPU over gloo (tools/twin_rank_check.py): the
    multi-rank engine flow — per-rank draws, fitness all-gather, redundant
    updates — yields bitwise-identical parameters on every rank."""
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--standalone", "--local-addr", "127.0.0.1",
           os.path.join(root, "tools", "twin_rank_check.py")]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                       env=dict(os.environ, PYTHONPATH=root))
    assert r.returncode == 0, r.stderr[-3000:]
    assert "TWIN-RANK OK" in r.stdout


def _pair_engine_fits(dev, std, pair, seed=92, gens=2, max_steps=20, ppg=8):
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    torch.manual_seed(seed)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": max_steps},
                    "noise": {"tbl_size": 1_000_000, "std": std},
                    "policy": {"layer_sizes": [64], "ac_std": 0.01, "l2coeff": 0.005,
                               "lr": 0.01, "ob_clip": 5, "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": ppg, "batch_size": 500,
                                "seed": 2}})
    env = make_batched("Humanoid-v2", ppg + 1, dev, max_steps=max_steps,
                       terminate_on_fall=True)
    nn = FeedForward([64], torch.nn.Tanh(), env, 0.01, 5)
    policy = Policy(nn, std, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=3, device=dev)
    rs = np.random.RandomState(seed + 1)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=True,
                    pair_rollout=pair)
    assert eng.pair_rollout == pair
    ranker = CenteredRanker()
    for _ in range(gens):
        eng.step(ranker)
    torch.cuda.synchronize(dev)
    fits = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
    return fits, eng.theta.cpu().clone(), eng.rew_total.cpu().clone()


def test_pair_rollout_sigma0_bitwise(dev):
    """sigma=0: pair effective weights bf16(theta)+-bf16(0) == the fused
    path's bf16(theta+0), so the whole pipeline (slot mapping, obs, decode,
    shared-A dynamics, epilogue) must match BITWISE."""
    import numpy as np
    a = _pair_engine_fits(dev, std=0.0, pair=False)
    b = _pair_engine_fits(dev, std=0.0, pair=True)
    np.testing.assert_array_equal(a[0], b[0])
    assert torch.equal(a[1], b[1]) and torch.equal(a[2], b[2])


def test_pair_rollout_sigma_close_and_deterministic(dev):
    """sigma>0: pair weights round twice (bf16(theta)+-bf16(sigma*eps)) vs
    once — short-horizon fitnesses stay close; the pair path is bitwise
    repeat-deterministic."""
    import numpy as np
    a = _pair_engine_fits(dev, std=0.02, pair=False, gens=1, max_steps=10)
    b = _pair_engine_fits(dev, std=0.02, pair=True, gens=1, max_steps=10)
    np.testing.assert_allclose(a[0], b[0], rtol=0.05, atol=0.5)
    c = _pair_engine_fits(dev, std=0.02, pair=True, gens=1, max_steps=10)
    np.testing.assert_array_equal(b[0], c[0])
    assert torch.equal(b[1], c[1])


def test_pair_rollout_variants_sigma0_bitwise(dev):
    """Pair path across slot-mapping variants — eps>1 (episode-averaged
    slots), goal-conditioned obs, odd S (scalar dynamics fallback) — each
    bitwise-equal to the fused path at sigma=0."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    for env_name, eps, kw in (("Humanoid-v2", 2, {}),
                              ("HumanoidFlagrunBulletEnv-v0", 1,
                               {"goal_conditioned": True}),
                              ("Walker2d-v3", 1, {})):
        out = {}
        for pair in (False, True):
            torch.manual_seed(77)
            comm = Comm(dev)
            cfg = AttrDict({"env": {"name": env_name, "max_steps": 15},
                            "noise": {"tbl_size": 600_000, "std": 0.0},
                            "policy": {"layer_sizes": [32], "ac_std": 0.01,
                                       "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                       "save_obs_chance": 1.0},
                            "general": {"policies_per_gen": 6, "batch_size": 500,
                                        "seed": 2, "eps_per_policy": eps}})
            B = 7 * eps
            env = make_batched(env_name, B, dev, max_steps=15, **kw)
            nn = FeedForward([32], torch.nn.Tanh(), env, 0.01, 5)
            policy = Policy(nn, 0.0, Adam(len(Policy.get_flat(nn)), 0.01))
            nt = NoiseTable.create_shared(comm, 600_000, len(policy), seed=3,
                                          device=dev)
            rs = np.random.RandomState(78)
            eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False,
                            pair_rollout=pair)
            assert eng.pair_rollout == pair, env_name
            ranker = CenteredRanker()
            eng.step(ranker)
            torch.cuda.synchronize(dev)
            out[pair] = (np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel(),
                         eng.rew_total.cpu().clone())
        np.testing.assert_array_equal(out[False][0], out[True][0], err_msg=env_name)
        assert torch.equal(out[False][1], out[True][1]), env_name


def test_pair_rollout_act_modes_sigma0_bitwise(dev):
    """Pair path with K9 binned decode and both integrated-gaussian modes:
    sigma=0 bitwise-equal to the fused path (the decode runs per slot in
    both, so the action noise/argmax indices must line up exactly)."""
    import numpy as np
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import (FFBinned, FFIntegGausAction,
                                      FFIntegGausActionMulti)
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm
    from es_pytorch_amd.utils.rankers import CenteredRanker

    from es_pytorch_amd.spaces import Box

    def build_net(kind, env):
        if kind == "binned":
            return FFBinned([32], torch.nn.Tanh(), env, n_bins=5, ob_clip=5)

        # unified output contract: these nets size their own output layer
        cls = FFIntegGausAction if kind == "ig" else FFIntegGausActionMulti
        return cls([32], torch.nn.Tanh(), env, ac_std=0.0, ob_clip=5)

    for kind in ("binned", "ig", "igm"):
        out = {}
        for pair in (False, True):
            torch.manual_seed(81)
            comm = Comm(dev)
            cfg = AttrDict({"env": {"name": "Hopper-v3", "max_steps": 15},
                            "noise": {"tbl_size": 500_000, "std": 0.0},
                            "policy": {"layer_sizes": [32], "ac_std": 0.0,
                                       "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                       "save_obs_chance": 1.0},
                            "general": {"policies_per_gen": 6, "batch_size": 500,
                                        "seed": 1}})
            env = make_batched("Hopper-v3", 7, dev, max_steps=15,
                               terminate_on_fall=False)
            nn = build_net(kind, env)
            policy = Policy(nn, 0.0, Adam(len(Policy.get_flat(nn)), 0.01))
            nt = NoiseTable.create_shared(comm, 500_000, len(policy), seed=4,
                                          device=dev)
            rs = np.random.RandomState(82)
            eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False,
                            pair_rollout=pair)
            assert eng.pair_rollout == pair
            ranker = CenteredRanker()
            eng.step(ranker)
            torch.cuda.synchronize(dev)
            out[pair] = np.concatenate([ranker.fits_pos, ranker.fits_neg]).ravel()
        np.testing.assert_array_equal(out[False], out[True], err_msg=kind)
