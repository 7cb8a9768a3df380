"""Policy HTTP serving (es_pytorch_amd/serve.py) via the ASGI test client."""
import numpy as np
import pytest
import torch

fastapi = pytest.importorskip("fastapi")
import warnings  # noqa: E402

with warnings.catch_warnings():
    # this image's fastapi/starlette pairing warns about its own TestClient
    # import; the deprecation is theirs, not ours
    warnings.simplefilter("ignore")
    from fastapi.testclient import TestClient  # noqa: E402

from es_pytorch_amd.core.policy import Policy  # noqa: E402
from es_pytorch_amd.envs import make  # noqa: E402
from es_pytorch_amd.nn.nn import FeedForward  # noqa: E402
from es_pytorch_amd.nn.optimizers import Adam  # noqa: E402
from es_pytorch_amd.serve import build_app, load_model  # noqa: E402


def _policy():
    torch.manual_seed(4)
    env = make("Hopper-v3")
    nn = FeedForward([16], torch.nn.Tanh(), env, ac_std=0.3, ob_clip=5)
    return Policy(nn, 0.02, Adam(len(Policy.get_flat(nn)), 0.01))


def test_serve_act_and_info(tmp_path):
    policy = _policy()
    policy.save(str(tmp_path), "7")
    client = TestClient(build_app(str(tmp_path / "policy-7")))

    assert client.get("/healthz").json() == {"ok": True}
    info = client.get("/info").json()
    assert info["ob_dim"] == 11 and info["ac_dim"] == 3
    assert info["n_params"] == len(policy)

    obs = [[0.1] * 11, [0.2] * 11]
    r = client.post("/act", json={"obs": obs})
    assert r.status_code == 200
    acts = np.asarray(r.json()["actions"])
    assert acts.shape == (2, 3)
    # deterministic (no action noise) and equal to the module's rs=None pass
    with torch.no_grad():
        ref = policy._module(torch.tensor(obs[0]), rs=None).numpy()
    np.testing.assert_allclose(acts[0], ref, rtol=1e-6)
    r2 = client.post("/act", json={"obs": obs})
    assert r2.json() == r.json()

    assert client.post("/act", json={"obs": []}).status_code == 422


def test_serve_torch_module(tmp_path):
    policy = _policy()
    torch.save(policy._module, tmp_path / "mod.pt")
    model = load_model(str(tmp_path / "mod.pt"))
    client = TestClient(build_app(model))
    r = client.post("/act", json={"obs": [[0.0] * 11]})
    assert r.status_code == 200 and len(r.json()["actions"][0]) == 3
