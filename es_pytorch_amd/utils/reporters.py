"""Per-generation observability: metric summaries, sinks, checkpoint policy.

API parity targets (class/method names only) are the reference's reporter
hierarchy, ``src/utils/reporters.py:26-270``; the implementation here is this
repo's own design: one pure function (:func:`generation_summary`) computes the
whole per-generation scoreboard, and thin sink classes decide where each
scalar/text line goes (stdout, a log file, MLflow). Rank gating uses the Comm
wrapper (torch.distributed) instead of mpi4py.

Semantics preserved because downstream tooling depends on them:

* metric keys/order (``avg-i``, ``max-i``, ``dist``, ``rew``, ``steps``,
  ``cum steps``, ``n fits ranked``, ``time``) — parsed by ``utils/viz.py``;
* only rank 0 emits or writes anything;
* ``DefaultReporterSet`` saves the policy on a new best reward OR distance and
  dumps each generation's fitness matrix under ``saved/<run>/fits/``.
"""
from __future__ import annotations

import logging
import os
import time
from abc import ABC, abstractmethod
from datetime import datetime
from pathlib import Path
from typing import Dict, List, Optional, Tuple

import numpy as np

from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import TrainingResult

#: sentinel emitted between metric groups (renders as a blank stdout line)
_BREAK = ("", None)


def calc_dist_rew(tr: TrainingResult) -> Tuple[float, float]:
    """(planar distance of the final position, total episode reward)."""
    x, y = tr.positions[-3], tr.positions[-2]
    return float(np.hypot(x, y)), float(np.sum(tr.rewards))


def generation_summary(fits: np.ndarray, noiseless_tr: TrainingResult,
                       steps: int, cum_steps: int,
                       ) -> List[Tuple[str, Optional[float]]]:
    """The whole per-generation scoreboard as an ordered (key, value) list.

    ``fits`` is the (pop, n_objectives) fitness matrix of every evaluated
    perturbation; ``noiseless_tr`` the unperturbed evaluation episode. A
    ``("", None)`` entry is a presentation break, not a metric.
    """
    cols = np.atleast_2d(np.asarray(fits, dtype=np.float64))
    board: List[Tuple[str, Optional[float]]] = []
    for i in range(cols.shape[1]):
        board.append((f"avg-{i}", float(np.round(cols[:, i].mean(), 2))))
        board.append((f"max-{i}", float(np.round(cols[:, i].max(), 2))))
    dist, rew = calc_dist_rew(noiseless_tr)
    board += [("dist", dist), ("rew", rew), _BREAK,
              ("steps", steps), ("cum steps", cum_steps),
              ("n fits ranked", len(cols))]
    return board


class Reporter(ABC):
    """Receiver of training progress; the engine/entry scripts call these."""

    @abstractmethod
    def start_gen(self): ...

    @abstractmethod
    def log_gen(self, fits: np.ndarray, noiseless_tr: TrainingResult, policy, steps: int): ...

    @abstractmethod
    def end_gen(self): ...

    @abstractmethod
    def print(self, s: str):
        """One-off informational text."""

    @abstractmethod
    def log(self, d: Dict[str, float]):
        """Recurring per-generation key/value pairs."""


class ReporterSet(Reporter):
    """Fan a Reporter call out to several receivers (None entries skipped)."""

    def __init__(self, *reporters: Optional[Reporter]):
        self.reporters = [r for r in reporters if r is not None]

    def _fan(self, method: str, *a):
        for r in self.reporters:
            getattr(r, method)(*a)

    def start_gen(self):
        self._fan("start_gen")

    def log_gen(self, fits, noiseless_tr, policy, steps):
        self._fan("log_gen", fits, noiseless_tr, policy, steps)

    def end_gen(self):
        self._fan("end_gen")

    def print(self, s: str):
        self._fan("print", s)

    def log(self, d: Dict[str, float]):
        self._fan("log", d)


class RankGatedReporter(Reporter, ABC):
    """Base for reporters that act on rank MAIN only and stay silent elsewhere.

    Subclasses implement the two sink hooks (``emit_text`` / ``emit_scalar``)
    plus optional lifecycle hooks; the public Reporter methods here do the
    gating, so non-main ranks pay one attribute check per call.
    """

    MAIN = 0

    def __init__(self, comm: Comm):
        self.comm = comm
        self._live = comm.rank == self.MAIN

    # -- sink hooks -------------------------------------------------------
    @abstractmethod
    def emit_text(self, s: str): ...

    @abstractmethod
    def emit_scalar(self, key: str, value: float): ...

    def on_gen_open(self):
        """Lifecycle hook: a generation is starting (rank MAIN only)."""

    def on_gen_close(self):
        """Lifecycle hook: a generation finished (rank MAIN only)."""

    def on_gen_board(self, fits, noiseless_tr, policy, steps):
        """Lifecycle hook: per-generation results available (rank MAIN only)."""

    # -- gated public API -------------------------------------------------
    def start_gen(self):
        if self._live:
            self.on_gen_open()

    def log_gen(self, fits, noiseless_tr, policy, steps):
        if self._live:
            self.on_gen_board(fits, noiseless_tr, policy, steps)

    def end_gen(self):
        if self._live:
            self.on_gen_close()

    def print(self, s: str):
        if self._live:
            self.emit_text(s)

    def log(self, d: Dict[str, float]):
        if self._live:
            for k, v in d.items():
                self.emit_scalar(k, v)


# name kept for reference-API familiarity (no MPI underneath)
MpiReporter = RankGatedReporter


class DefaultReporter(RankGatedReporter, ABC):
    """Adds the standard scoreboard + gen/step counters + wall-clock timing."""

    def __init__(self, comm: Comm):
        super().__init__(comm)
        self.gen = 0
        self.cum_steps = 0
        self._opened_at = 0.0

    def on_gen_open(self):
        self._opened_at = time.time()
        self.emit_text("\n\n----------------------------------------")
        self.emit_scalar("gen", self.gen)

    def on_gen_board(self, fits, noiseless_tr, policy, steps):
        self.cum_steps += steps
        for key, val in generation_summary(fits, noiseless_tr, steps, self.cum_steps):
            if (key, val) == _BREAK:
                self.emit_text("")
            else:
                self.emit_scalar(key, val)

    def on_gen_close(self):
        self.emit_scalar("time", round(time.time() - self._opened_at, 2))
        self.gen += 1


DefaultMpiReporter = DefaultReporter


class DefaultReporterSet(DefaultReporter):
    """Scoreboard + artifact policy, fanned out to child reporters.

    Owns the run directory layout ``<save_root>/<run_name>/{fits,weights}``:
    every generation's fitness matrix is dumped as ``fits/<gen>.np.npy``, and
    the policy is checkpointed to ``weights/`` whenever the noiseless episode
    sets a new best total reward or planar distance (either suffices —
    locomotion runs often improve distance before reward).
    """

    def __init__(self, comm: Comm, run_name: str, *reporters: Optional[Reporter],
                 save_root: str = "saved"):
        super().__init__(comm)
        self.reporters = [r for r in reporters if r is not None]
        root = Path(save_root) / run_name
        self.fit_folder = str(root / "fits")
        self.policy_folder = str(root / "weights")
        if self._live:
            for d in (self.fit_folder, self.policy_folder):
                os.makedirs(d, exist_ok=True)
        self.best_rew = 0.0
        self.best_dist = 0.0

    def emit_text(self, s: str):
        for r in self.reporters:
            r.print(s)

    def emit_scalar(self, key: str, value: float):
        for r in self.reporters:
            r.log({key: value})

    def on_gen_board(self, fits, noiseless_tr, policy, steps):
        super().on_gen_board(fits, noiseless_tr, policy, steps)
        dist, rew = calc_dist_rew(noiseless_tr)
        if rew > self.best_rew or dist > self.best_dist:
            policy.save(self.policy_folder, str(self.gen))
            self.emit_text(f"saving policy with rew:{rew:0.2f} and dist:{dist:0.2f}")
        self.best_rew = max(self.best_rew, rew)
        self.best_dist = max(self.best_dist, dist)
        np.save(os.path.join(self.fit_folder, f"{self.gen}.np"), np.asarray(fits))


DefaultMpiReporterSet = DefaultReporterSet


class StdoutReporter(DefaultReporter):
    """Plain ``key:value`` lines on stdout."""

    def emit_text(self, s: str):
        print(s)

    def emit_scalar(self, key: str, value: float):
        print(f"{key}:{value}")


class LoggerReporter(DefaultReporter):
    """Same stream, through the stdlib logging module into
    ``<save_root>/<log_folder>/es.log`` (folder defaults to a timestamp)."""

    def __init__(self, comm: Comm, log_folder: Optional[str] = None,
                 save_root: str = "saved"):
        super().__init__(comm)
        if self._live:
            folder = log_folder or datetime.now().strftime("es__%d_%m_%y__%H_%M_%S")
            os.makedirs(os.path.join(save_root, folder), exist_ok=True)
            logging.basicConfig(
                filename=os.path.join(save_root, folder, "es.log"),
                level=logging.DEBUG)
            logging.info("initialized logger")

    def emit_text(self, s: str):
        logging.info(s)

    def emit_scalar(self, key: str, value: float):
        logging.info(f"{key}:{value}")


class MLFlowReporter(DefaultReporter):
    """MLflow sink: one parent run for the experiment, one nested run per
    population member (multi-policy novelty search trains several policies;
    each gets its own metric timeline). ``set_active_run(i)`` selects which
    nested run the coming generation's metrics land in.

    Requires the optional ``mlflow`` package; constructing this without it
    raises ImportError (loud, not silent)."""

    def __init__(self, comm: Comm, cfg):
        super().__init__(comm)
        import mlflow  # optional dependency: fail at construction, not mid-run
        self._mlflow = mlflow
        self.active_run: Optional[int] = None
        if self._live:
            mlflow.set_experiment(cfg.env.name)
            mlflow.start_run(run_name=cfg.general.name)
            mlflow.log_params(_flatten_cfg(cfg.to_dict()))
            n = cfg.general.n_policies
            self.gens = [0] * n
            self.run_ids = [self._open_nested(str(i)) for i in range(n)]

    def _open_nested(self, name: str) -> str:
        with self._mlflow.start_run(run_name=name, nested=True) as run:
            return run.info.run_id

    def set_active_run(self, i: int):
        if self._live:
            self.active_run = i

    def start_active_run(self):
        assert self.active_run is not None, "call set_active_run first"
        return self._mlflow.start_run(run_id=self.run_ids[self.active_run],
                                      nested=True)

    def on_gen_open(self):
        pass  # no per-gen banner in MLflow

    def on_gen_close(self):
        self.gens[self.active_run] += 1
        self.active_run = None

    def emit_text(self, s: str):
        pass  # free-form text has no MLflow home

    def emit_scalar(self, key: str, value: float):
        with self.start_active_run():
            self._mlflow.log_metrics({key: value}, self.gens[self.active_run])


def _flatten_cfg(tree: dict, prefix: str = "") -> dict:
    """Nested config dict -> dotted flat dict (MLflow params are flat)."""
    flat = {}
    for k, v in tree.items():
        if isinstance(v, dict):
            flat.update(_flatten_cfg(v, f"{prefix}{k}."))
        else:
            flat[f"{prefix}{k}"] = v
    return flat
