"""MLFlowReporter logging path, exercised against a stub mlflow backend.

mlflow is not installed offline; the reporter imports it at construction
(the seam), so a stub module in sys.modules lets the full
set_active_run -> log_gen -> end_gen flow run and be asserted on
(reference behaviour: ``src/utils/reporters.py:232-270`` — one nested run
per population member, metrics stepped by that member's own gen counter)."""
import sys
import types

import numpy as np
import pytest
import torch

from es_pytorch_amd.config import AttrDict
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import RewardResult


class _StubRun:
    def __init__(self, run_id, name, nested):
        self.info = types.SimpleNamespace(run_id=run_id)
        self.name, self.nested = name, nested

    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


class _StubMlflow(types.ModuleType):
    def __init__(self):
        super().__init__("mlflow")
        self.experiments = []
        self.params = {}
        self.metrics = []  # (run_id, key, value, step)
        self._runs = {}
        self._stack = []
        self._next = 0

    def set_experiment(self, name):
        self.experiments.append(name)

    def start_run(self, run_name=None, run_id=None, nested=False):
        if run_id is None:
            run_id = f"run{self._next}"
            self._next += 1
        run = self._runs.setdefault(run_id, _StubRun(run_id, run_name, nested))
        self._stack.append(run)
        return run

    def log_params(self, d):
        self.params.update(d)

    def log_metrics(self, d, step):
        rid = self._stack[-1].info.run_id if self._stack else None
        for k, v in d.items():
            self.metrics.append((rid, k, v, step))


@pytest.fixture
def stub_mlflow(monkeypatch):
    stub = _StubMlflow()
    monkeypatch.setitem(sys.modules, "mlflow", stub)
    return stub


def _cfg(n_policies=2):
    return AttrDict({"env": {"name": "Hopper-v2"},
                     "general": {"name": "testrun", "n_policies": n_policies},
                     "noise": {"std": 0.02}})


def test_mlflow_reporter_full_flow(stub_mlflow):
    from es_pytorch_amd.utils.reporters import MLFlowReporter
    comm = Comm(torch.device("cpu"))
    rep = MLFlowReporter(comm, _cfg(n_policies=2))

    # construction: experiment + parent run + config params + 2 nested runs
    assert stub_mlflow.experiments == ["Hopper-v2"]
    assert stub_mlflow.params["general.name"] == "testrun"
    assert stub_mlflow.params["noise.std"] == 0.02
    assert len(rep.run_ids) == 2

    tr = RewardResult([1.0, 2.0], [0.5, 0.5, 0.1] * 2, np.zeros((1, 3)), 2)
    fits = np.array([[1.0], [2.0]])

    rep.set_active_run(1)
    rep.start_gen()
    rep.log_gen(fits, tr, policy=None, steps=4)
    rep.end_gen()

    member1 = rep.run_ids[1]
    logged = {(k, step) for rid, k, v, step in stub_mlflow.metrics if rid == member1}
    assert ("avg-0", 0) in logged and ("rew", 0) in logged and ("steps", 0) in logged
    assert rep.gens[1] == 1 and rep.active_run is None

    # second generation for the same member logs at step 1
    rep.set_active_run(1)
    rep.start_gen()
    rep.log_gen(fits, tr, policy=None, steps=4)
    rep.end_gen()
    assert ("rew", 1) in {(k, s) for rid, k, v, s in stub_mlflow.metrics
                          if rid == member1}
    # member 0's timeline untouched
    assert all(rid != rep.run_ids[0] for rid, *_ in stub_mlflow.metrics)


def test_mlflow_requires_active_run(stub_mlflow):
    from es_pytorch_amd.utils.reporters import MLFlowReporter
    rep = MLFlowReporter(Comm(torch.device("cpu")), _cfg(1))
    with pytest.raises(AssertionError, match="set_active_run"):
        rep.log({"x": 1.0})


def test_mlflow_import_error_without_package():
    """No stub, no mlflow installed -> loud ImportError at construction."""
    assert "mlflow" not in sys.modules or not hasattr(sys.modules.get("mlflow"), "__file__")
    from es_pytorch_amd.utils.reporters import MLFlowReporter
    if "mlflow" in sys.modules:
        pytest.skip("a stub/real mlflow is importable in this session")
    with pytest.raises(ImportError):
        MLFlowReporter(Comm(torch.device("cpu")), _cfg(1))
