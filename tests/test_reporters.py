"""Reporter hierarchy unit tests (reference src/utils/reporters.py semantics)."""
import os

import numpy as np
import torch

from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import RewardResult
from es_pytorch_amd.utils.reporters import (DefaultReporterSet, Reporter, ReporterSet,
                                            StdoutReporter, calc_dist_rew)


class _Capture(Reporter):
    def __init__(self):
        self.lines, self.logs = [], []

    def start_gen(self):
        pass

    def log_gen(self, fits, tr, policy, steps):
        pass

    def end_gen(self):
        pass

    def print(self, s):
        self.lines.append(s)

    def log(self, d):
        self.logs.append(dict(d))


def _tr(rews=(1.0, 2.0), pos=(3.0, 4.0, 0.5, 3.0, 4.0, 0.5)):
    return RewardResult(list(rews), list(pos), np.zeros((1, 2)), len(rews))


def test_calc_dist_rew():
    dist, rew = calc_dist_rew(_tr())
    assert abs(dist - 5.0) < 1e-6  # norm of final (x, y) = (3, 4)
    assert rew == 3.0


def test_reporter_set_fanout():
    a, b = _Capture(), _Capture()
    rs = ReporterSet(a, None, b)  # None entries dropped (reference reporters.py:54)
    rs.print("hello")
    rs.log({"k": 1})
    assert a.lines == b.lines == ["hello"]
    assert a.logs == b.logs == [{"k": 1}]


def test_stdout_reporter_gen_metrics(capsys):
    comm = Comm(torch.device("cpu"))
    r = StdoutReporter(comm)
    r.start_gen()
    fits = np.array([[1.0], [3.0], [2.0], [4.0]])
    r.log_gen(fits, _tr(), policy=None, steps=10)
    r.end_gen()
    out = capsys.readouterr().out
    assert "avg-0:2.5" in out
    assert "max-0:4.0" in out
    assert "steps:10" in out
    assert "cum steps:10" in out
    assert r.gen == 1


class _FakePolicy:
    def __init__(self):
        self.saved = []

    def save(self, folder, suffix):
        os.makedirs(folder, exist_ok=True)
        self.saved.append(suffix)


def test_default_reporter_set_saves_on_best(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    comm = Comm(torch.device("cpu"))
    cap = _Capture()
    rset = DefaultReporterSet(comm, "run1", cap)
    pol = _FakePolicy()
    fits = np.array([[1.0], [2.0]])
    rset.start_gen()
    rset.log_gen(fits, _tr(rews=(5.0,)), pol, steps=1)  # rew 5 > best 0 -> save
    rset.end_gen()
    rset.start_gen()
    rset.log_gen(fits, _tr(rews=(1.0,)), pol, steps=1)  # worse -> no save
    rset.end_gen()
    assert pol.saved == ["0"]
    # per-gen fitness dumps written (reference reporters.py:188)
    assert os.path.exists(tmp_path / "saved" / "run1" / "fits" / "0.np.npy")
    assert os.path.exists(tmp_path / "saved" / "run1" / "fits" / "1.np.npy")
