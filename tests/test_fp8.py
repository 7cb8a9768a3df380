"""fp8 (OCP e4m3fn) sigma*eps stream: layout, encoding and fidelity.

The fp8 path halves the pair rollout's dominant HBM stream; these tests pin
(a) the pheno_fp8 blob's row-pair-interleaved layout + e4m3fn encoding
against an independent numpy model, (b) plumbing correctness at sigma=0
(same trajectories as the bf16 pair path up to summation-order rounding),
and (c) rank/gradient fidelity at the flagship sigma.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda", 0)


def e4m3fn_decode(b: np.ndarray) -> np.ndarray:
    """OCP e4m3fn byte -> float (numpy; NaN at 0x7F/0xFF)."""
    b = b.astype(np.uint32)
    sign = np.where(b & 0x80, -1.0, 1.0)
    exp = (b >> 3) & 0xF
    man = b & 0x7
    val = np.where(exp == 0, (man / 8.0) * 2.0 ** -6,
                   (1.0 + man / 8.0) * 2.0 ** (exp.astype(np.int32) - 7))
    val = np.where((exp == 15) & (man == 7), np.nan, val)
    return sign * val


def _interleave_map(dims, n):
    """element index -> byte position in the fp8 row (pheno.hip layout)."""
    pos = np.arange(n, dtype=np.int64)
    out = pos.copy()
    off = 0
    for I, O in zip(dims[:-1], dims[1:]):
        vec = (O % 8 == 0) and (off % 8 == 0)
        if vec:
            ro = I % 2  # odd input dim: row 0 stays plain
            e = np.arange(ro * O, I * O, dtype=np.int64)
            i, o = e // O - ro, e % O
            byte = ro * O + (i & ~1) * O + (o >> 3) * 16 + (i & 1) * 8 + (o & 7)
            out[off + ro * O: off + I * O] = off + byte
        off += I * O + O  # bias block stays identity
    return out


def _mk(dev, pop=64, steps=20, std=0.02, fp8=False, layers=(64, 64), eps=1,
        env_name="Humanoid-v2"):
    from es_pytorch_amd.config import AttrDict
    from es_pytorch_amd.core.engine import GpuEngine
    from es_pytorch_amd.core.noisetable import NoiseTable
    from es_pytorch_amd.core.policy import Policy
    from es_pytorch_amd.envs import make_batched
    from es_pytorch_amd.nn.nn import FeedForward
    from es_pytorch_amd.nn.optimizers import Adam
    from es_pytorch_amd.parallel.comm import Comm

    torch.manual_seed(21)
    comm = Comm(dev)
    cfg = AttrDict({"env": {"name": env_name, "max_steps": steps},
                    "noise": {"tbl_size": 2_000_000, "std": std},
                    "policy": {"layer_sizes": list(layers), "ac_std": 0.0,
                               "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                               "save_obs_chance": 1.0},
                    "general": {"policies_per_gen": pop, "batch_size": 500,
                                "seed": 3, "eps_per_policy": eps}})
    env = make_batched(env_name, (pop + 1) * eps, dev, max_steps=steps,
                       terminate_on_fall=False)
    nn = FeedForward(list(layers), torch.nn.Tanh(), env, 0.0, 5)
    policy = Policy(nn, std, Adam(len(Policy.get_flat(nn)), 0.01))
    nt = NoiseTable.create_shared(comm, 2_000_000, len(policy), seed=7, device=dev)
    rs = np.random.RandomState(5)
    eng = GpuEngine(cfg, comm, policy, nt, env, rs, use_graph=False,
                    pair_rollout=True, eps_fp8=fp8)
    return eng


@pytest.mark.parametrize("env_name,layers", [
    ("Humanoid-v2", (64, 64)),
    # odd input dim (AntFlagrun obs = 29): row 0 plain, rows 1.. paired
    ("AntFlagrunBulletEnv-v0", (64, 64)),
])
def test_pheno_fp8_layout_and_encoding(dev, env_name, layers):
    """The fp8 blob, de-interleaved and decoded on the host, must equal
    e4m3fn(sigma * table[offset + t]) elementwise."""
    eng = _mk(dev, fp8=True, layers=layers, env_name=env_name)
    assert eng.eps_fp8, f"fp8 gate must accept {env_name}"
    eng._upload_offsets()
    eng._pheno()
    torch.cuda.synchronize(dev)

    blob = eng.eps_rows.cpu().numpy()          # (pairs, row_stride) uint8
    table = eng.nt.noise.cpu().numpy()
    offs = eng.offsets.cpu().numpy()
    m = _interleave_map(eng.dims, eng.n)
    std = float(eng.policy.std)
    for p_i in [0, eng.pairs - 1]:
        want = std * table[offs[p_i]: offs[p_i] + eng.n]
        got = e4m3fn_decode(blob[p_i][m])       # de-interleave + decode
        assert not np.isnan(got).any()
        # e4m3 RNE: rel err <= 2^-4 for normals, abs floor for subnormals
        err = np.abs(got - want)
        tol = np.maximum(np.abs(want) * 0.0625, 2.0 ** -7)
        bad = err > tol
        assert not bad.any(), (p_i, np.argwhere(bad)[:5], want[bad][:5], got[bad][:5])


@pytest.mark.parametrize("eps,env_name", [(1, "Humanoid-v2"), (2, "Humanoid-v2"),
                                          (1, "AntFlagrunBulletEnv-v0")])
def test_fp8_sigma0_matches_bf16_pair(dev, eps, env_name):
    """At sigma=0 both eps formats encode exact zeros; trajectories agree up
    to the fp8 path's different per-thread summation partition (also with
    episode averaging, eps_per_policy=2)."""
    outs = {}
    for fp8 in (False, True):
        eng = _mk(dev, std=0.0, fp8=fp8, eps=eps, env_name=env_name)
        if env_name != "Humanoid-v2":
            assert eng.eps_fp8 == fp8
        from es_pytorch_amd.utils.rankers import CenteredRanker
        eng.step(CenteredRanker())
        torch.cuda.synchronize(dev)
        outs[fp8] = eng.rew_total.cpu().numpy().copy()
    np.testing.assert_allclose(outs[True], outs[False], rtol=2e-4, atol=2e-3)


def test_quantized_gather_matches_rollout_perturbation(dev):
    """es_grad_gather(qstd=sigma) must use EXACTLY the perturbation values
    the fp8 rollout evaluated: with a single unit fitness, sigma * g equals
    the decoded pheno_fp8 blob elementwise (same hardware converters on
    both paths -> estimator-exact ES on the quantized distribution)."""
    from es_pytorch_amd import ops
    eng = _mk(dev, fp8=True)
    eng._upload_offsets()
    eng._pheno()
    torch.cuda.synchronize(dev)
    std = float(eng.policy.std)

    fits = torch.ones(1, dtype=torch.float32, device=dev)
    offs = eng.offsets[:1].contiguous()
    g = torch.empty(eng.n, dtype=torch.float32, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    ops.check(ops.hip().es_grad_gather(g.data_ptr(), eng.nt.noise.data_ptr(),
                                       fits.data_ptr(), offs.data_ptr(), 1, eng.n,
                                       std, stream), "grad_q")
    torch.cuda.synchronize(dev)

    blob = eng.eps_rows[0].cpu().numpy()
    decoded = e4m3fn_decode(blob[_interleave_map(eng.dims, eng.n)]).astype(np.float64)
    np.testing.assert_allclose(std * g.cpu().numpy().astype(np.float64), decoded,
                               rtol=1e-6, atol=1e-12)


@pytest.mark.parametrize("kind", ["binned", "ig", "igm"])
def test_fp8_act_modes_sigma0(dev, kind):
    """fp8 with the K9 binned decode and both integrated-gaussian heads:
    sigma=0 must agree with the bf16 pair path (the non-8-aligned heads run
    the scalar fp8 layout — plain element-ordered bytes)."""
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

    outs = {}
    for fp8 in (False, True):
        torch.manual_seed(31)
        comm = Comm(dev)
        cfg = AttrDict({"env": {"name": "Humanoid-v2", "max_steps": 15},
                        "noise": {"tbl_size": 1_000_000, "std": 0.0},
                        "policy": {"layer_sizes": [32], "ac_std": 0.0,
                                   "l2coeff": 0.005, "lr": 0.01, "ob_clip": 5,
                                   "save_obs_chance": 1.0},
                        "general": {"policies_per_gen": 8, "batch_size": 500,
                                    "seed": 6}})
        env = make_batched("Humanoid-v2", 9, dev, max_steps=15,
                           terminate_on_fall=False)
        if kind == "binned":
            nn = FFBinned([32], torch.nn.Tanh(), env, n_bins=5, ob_clip=5)
        else:
            cls = FFIntegGausAction if kind == "ig" else FFIntegGausActionMulti
            nn = cls([32], torch.nn.Tanh(), env, ac_std=0.0, ob_clip=5)
        policy = Policy(nn, 0.0, Adam(len(Policy.get_flat(nn)), 0.01))
        nt = NoiseTable.create_shared(comm, 1_000_000, len(policy), seed=9,
                                      device=dev)
        eng = GpuEngine(cfg, comm, policy, nt, env, np.random.RandomState(7),
                        use_graph=False, pair_rollout=True, eps_fp8=fp8)
        if fp8:
            assert eng.eps_fp8, kind
        eng.step(CenteredRanker())
        torch.cuda.synchronize(dev)
        outs[fp8] = eng.rew_total.cpu().numpy().copy()
    np.testing.assert_allclose(outs[True], outs[False], rtol=2e-4, atol=2e-3,
                               err_msg=kind)


def test_fp8_fidelity_at_flagship_sigma(dev):
    """sigma=0.02, 200 steps, pop 512: member fitness ranking and the
    reconstructed gradient must track the bf16 pair path closely."""
    from scipy.stats import spearmanr

    from es_pytorch_amd.utils.rankers import CenteredRanker
    res = {}
    for fp8 in (False, True):
        eng = _mk(dev, pop=512, steps=200, std=0.02, fp8=fp8, layers=(256, 256))
        r = CenteredRanker()
        eng.step(r)
        torch.cuda.synchronize(dev)
        res[fp8] = (np.concatenate([r.fits_pos, r.fits_neg]).ravel(),
                    eng.grad.cpu().numpy().copy())
    rho = spearmanr(res[False][0], res[True][0]).correlation
    g0, g1 = res[False][1], res[True][1]
    cos = float(np.dot(g0, g1) / (np.linalg.norm(g0) * np.linalg.norm(g1)))
    print(f"fp8 fidelity: spearman {rho:.4f} cosine {cos:.4f}")
    assert rho > 0.9, rho
    assert cos > 0.9, cos
