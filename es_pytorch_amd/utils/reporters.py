"""Observability + checkpoint policy.

Same hierarchy as the reference (``src/utils/reporters.py:26-270``):
``Reporter`` ABC, composition via ``ReporterSet`` / ``DefaultMpiReporterSet``,
rank-0-only gating (``reporters.py:84-102``), per-generation metrics
(avg/max per objective, dist, rew, steps, cum steps, wall time,
``reporters.py:140-158``), best-policy checkpointing + per-gen fitness dumps
to ``saved/<run>/`` (``reporters.py:177-188``), python-logging file sink
(``reporters.py:211-229``) and an optional MLflow sink (``reporters.py:232-
270``; active only if mlflow is importable).

Rank gating uses the Comm wrapper (torch.distributed) instead of mpi4py.
"""
from __future__ import annotations

import logging
import os
import time
from abc import ABC, abstractmethod
from datetime import datetime
from os import path
from typing import Dict, Tuple

import numpy as np

from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import TrainingResult


def calc_dist_rew(tr: TrainingResult) -> Tuple[float, float]:
    """Distance traveled (ignoring height) and total reward (reference ``reporters.py:21-23``)."""
    return float(np.linalg.norm(np.array(tr.positions[-3:-1]))), float(np.sum(tr.rewards))


class Reporter(ABC):
    @abstractmethod
    def start_gen(self):
        ...

    @abstractmethod
    def log_gen(self, fits: np.ndarray, noiseless_tr: TrainingResult, policy, steps: int):
        ...

    @abstractmethod
    def end_gen(self):
        ...

    @abstractmethod
    def print(self, s: str):
        """One-time information."""

    @abstractmethod
    def log(self, d: Dict[str, float]):
        """Recurring per-generation key/value pairs."""


class ReporterSet(Reporter):
    def __init__(self, *reporters: Reporter):
        self.reporters = [r for r in reporters if r is not None]

    def start_gen(self):
        for r in self.reporters:
            r.start_gen()

    def log_gen(self, fits, noiseless_tr, policy, steps):
        for r in self.reporters:
            r.log_gen(fits, noiseless_tr, policy, steps)

    def end_gen(self):
        for r in self.reporters:
            r.end_gen()

    def print(self, s: str):
        for r in self.reporters:
            r.print(s)

    def log(self, d: Dict[str, float]):
        for r in self.reporters:
            r.log(d)


class RankGatedReporter(Reporter, ABC):
    """Rank-0-only reporter (reference ``MpiReporter``, ``reporters.py:77-122``)."""

    MAIN = 0

    def __init__(self, comm: Comm):
        self.comm = comm

    def start_gen(self):
        if self.comm.rank == self.MAIN:
            self._start_gen()

    def log_gen(self, fits, noiseless_tr, policy, steps):
        if self.comm.rank == self.MAIN:
            self._log_gen(fits, noiseless_tr, policy, steps)

    def end_gen(self):
        if self.comm.rank == self.MAIN:
            self._end_gen()

    def print(self, s: str):
        if self.comm.rank == self.MAIN:
            self._print(s)

    def log(self, d: Dict[str, float]):
        if self.comm.rank == self.MAIN:
            self._log(d)

    @abstractmethod
    def _start_gen(self): ...

    @abstractmethod
    def _log_gen(self, fits, noiseless_tr, policy, steps): ...

    @abstractmethod
    def _end_gen(self): ...

    @abstractmethod
    def _print(self, s: str): ...

    @abstractmethod
    def _log(self, d: Dict[str, float]): ...


# reference-compatible alias
MpiReporter = RankGatedReporter


class DefaultReporter(RankGatedReporter, ABC):
    """Standard per-gen metrics (reference ``DefaultMpiReporter``, ``reporters.py:125-159``)."""

    def __init__(self, comm: Comm):
        super().__init__(comm)
        self.gen = 0
        self.cum_steps = 0
        self.gen_start_time = 0.0

    def _start_gen(self):
        self.gen_start_time = time.time()
        self.print("\n\n----------------------------------------")
        self.log({"gen": self.gen})

    def _log_gen(self, fits: np.ndarray, noiseless_tr: TrainingResult, policy, steps: int):
        fits = np.atleast_2d(np.asarray(fits))
        for i, col in enumerate(fits.T):
            self.log({f"avg-{i}": float(np.round(np.mean(col), 2))})
            self.log({f"max-{i}": float(np.round(np.max(col), 2))})

        self.cum_steps += steps
        dist, rew = calc_dist_rew(noiseless_tr)
        self.log({"dist": dist})
        self.log({"rew": rew})
        self.print("")
        self.log({"steps": steps})
        self.log({"cum steps": self.cum_steps})
        self.log({"n fits ranked": len(fits)})

    def _end_gen(self):
        self.log({"time": round(time.time() - self.gen_start_time, 2)})
        self.gen += 1


DefaultMpiReporter = DefaultReporter


class DefaultReporterSet(DefaultReporter):
    """Composition + best-policy checkpointing + fitness dumps
    (reference ``DefaultMpiReporterSet``, ``reporters.py:162-196``)."""

    def __init__(self, comm: Comm, run_name: str, *reporters: Reporter, save_root: str = "saved"):
        super().__init__(comm)

        self.fit_folder = path.join(save_root, run_name, "fits")
        self.policy_folder = path.join(save_root, run_name, "weights")
        if comm.rank == self.MAIN:
            os.makedirs(self.fit_folder, exist_ok=True)
            os.makedirs(self.policy_folder, exist_ok=True)

        self.reporters = [r for r in reporters if r is not None]
        self.best_rew = 0.0
        self.best_dist = 0.0

    def _log_gen(self, fits, noiseless_tr, policy, steps):
        super()._log_gen(fits, noiseless_tr, policy, steps)
        dist, rew = calc_dist_rew(noiseless_tr)
        save_policy = (rew > self.best_rew or dist > self.best_dist)
        self.best_rew = max(rew, self.best_rew)
        self.best_dist = max(dist, self.best_dist)
        if save_policy:
            policy.save(self.policy_folder, str(self.gen))
            self.print(f"saving policy with rew:{rew:0.2f} and dist:{dist:0.2f}")
        np.save(path.join(self.fit_folder, f"{self.gen}.np"), np.asarray(fits))

    def _log(self, d):
        for r in self.reporters:
            r.log(d)

    def _print(self, s):
        for r in self.reporters:
            r.print(s)


DefaultMpiReporterSet = DefaultReporterSet


class StdoutReporter(DefaultReporter):
    def _print(self, s: str):
        print(s)

    def _log(self, d: Dict[str, float]):
        for k, v in d.items():
            print(f"{k}:{v}")


class LoggerReporter(DefaultReporter):
    """Python-logging file sink (reference ``reporters.py:211-229``)."""

    def __init__(self, comm: Comm, log_folder=None, save_root: str = "saved"):
        super().__init__(comm)
        if comm.rank == self.MAIN:
            if log_folder is None:
                log_folder = datetime.now().strftime("es__%d_%m_%y__%H_%M_%S")
            os.makedirs(path.join(save_root, log_folder), exist_ok=True)
            logging.basicConfig(filename=path.join(save_root, log_folder, "es.log"),
                                level=logging.DEBUG)
            logging.info("initialized logger")

    def _print(self, s: str):
        logging.info(s)

    def _log(self, d: Dict[str, float]):
        for k, v in d.items():
            logging.info(f"{k}:{v}")


class MLFlowReporter(DefaultReporter):
    """MLflow sink with one nested run per population member
    (reference ``reporters.py:232-270``). Requires the optional mlflow package."""

    def __init__(self, comm: Comm, cfg):
        super().__init__(comm)
        import mlflow  # noqa: optional dependency, fail loudly only if used
        self._mlflow = mlflow
        if comm.rank == self.MAIN:
            mlflow.set_experiment(cfg.env.name)
            mlflow.start_run(run_name=cfg.general.name)
            flat = _flatten_cfg(cfg.to_dict())
            mlflow.log_params(flat)
            self.gens = [0] * cfg.general.n_policies
            self.run_ids = []
            self.active_run = None
            for i in range(cfg.general.n_policies):
                with mlflow.start_run(run_name=f"{i}", nested=True) as run:
                    self.run_ids.append(run.info.run_id)

    def set_active_run(self, i: int):
        if self.comm.rank == self.MAIN:
            self.active_run = i

    def start_active_run(self):
        assert self.active_run is not None, "call set_active_run first"
        return self._mlflow.start_run(run_id=self.run_ids[self.active_run], nested=True)

    def _start_gen(self):
        pass

    def _end_gen(self):
        self.gens[self.active_run] += 1
        self.active_run = None

    def _print(self, s: str):
        pass

    def _log(self, d: Dict[str, float]):
        with self.start_active_run():
            self._mlflow.log_metrics(d, self.gens[self.active_run])


def _flatten_cfg(d: dict, prefix: str = "") -> dict:
    out = {}
    for k, v in d.items():
        key = f"{prefix}{k}"
        if isinstance(v, dict):
            out.update(_flatten_cfg(v, f"{key}."))
        else:
            out[key] = v
    return out
