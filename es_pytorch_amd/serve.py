"""Policy serving: a minimal HTTP inference endpoint for trained policies.

Beyond reference parity (the reference only replays policies with a local
renderer, ``run_saved.py``): wraps a saved policy — a whole-``Policy`` pickle
(``policy-<gen>``, core/policy.py format) or a ``torch.save``d module — in a
FastAPI app for deployment:

* ``GET  /info``    — obs/action dims, parameter count, source path
* ``GET  /healthz`` — liveness
* ``POST /act``     — ``{"obs": [[...], ...]}`` (one or many observation
  rows) → ``{"actions": [[...], ...]}``; deterministic (no action noise)

Run: ``python tools/serve_policy.py saved/<run>/weights/policy-40 --port 8080``
"""
import os
import pickle
import numpy as np
import torch


def load_model(path: str) -> torch.nn.Module:
    """A saved ``Policy`` pickle or a ``torch.save``d module -> eval module."""
    with open(path, "rb") as f:
        head = f.read(2)
    if head == b"PK":  # zipfile -> torch.save archive
        model = torch.load(path, weights_only=False)
    else:
        from es_pytorch_amd.core.policy import Policy
        with open(path, "rb") as f:
            obj = pickle.load(f)
        if isinstance(obj, Policy):
            obj.set_nn_params(obj.flat_params)
            model = obj._module
        else:
            model = obj
    model.eval()
    return model


def build_app(model_or_path):
    from fastapi import FastAPI, HTTPException, Request

    model = load_model(model_or_path) if isinstance(model_or_path, str) \
        else model_or_path
    model.eval()
    src = model_or_path if isinstance(model_or_path, str) else type(model).__name__

    app = FastAPI(title="es_pytorch_amd policy server")

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    @app.get("/info")
    def info():
        n_params = int(sum(p.numel() for p in model.parameters()))
        dims = model.layer_dims() if hasattr(model, "layer_dims") else None
        return {"source": os.path.basename(str(src)), "n_params": n_params,
                "ob_dim": dims[0] if dims else None,
                "ac_dim": dims[-1] if dims else None}

    @app.post("/act")
    async def act(request: Request):
        try:
            obs = np.asarray((await request.json())["obs"], dtype=np.float32)
        except (KeyError, TypeError, ValueError):
            raise HTTPException(422, "body must be {\"obs\": [[...], ...]}")
        if obs.ndim != 2 or obs.shape[0] == 0:
            raise HTTPException(422, "obs must be a non-empty list of rows")
        with torch.no_grad():
            acts = [np.asarray(model(torch.from_numpy(row), rs=None))
                    for row in obs]  # rs=None -> deterministic, no action noise
        return {"actions": [a.tolist() for a in acts]}

    return app
