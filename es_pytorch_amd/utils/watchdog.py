"""Failure detection for long runs: a generation watchdog + rank heartbeats.

The reference has no failure handling at all (SURVEY.md §5 aux gap); the
engine already fails loudly on non-finite parameters (core/engine.py
sync_host). This module covers the other two production failure modes:

* **Hangs** — a wedged collective (one rank dead, the rest blocked in RCCL)
  or a stuck kernel leaves the job silently burning GPU-hours.
  :class:`Watchdog` arms a timer around each generation; if the generation
  does not complete in time it dumps every thread's stack (faulthandler) and
  hard-exits the process, turning an invisible hang into a loud, restartable
  failure (pair with utils/checkpoint.py's ring for automatic resume).
* **Silent rank death** — :class:`Heartbeat` has every rank write an atomic
  per-rank beat file each generation; an external monitor (or any other
  rank's operator) calls :meth:`Heartbeat.stalled_ranks` to see who stopped
  making progress and how far behind they are.
"""
from __future__ import annotations

import faulthandler
import json
import os
import sys
import threading
import time
from typing import Callable, Dict, List, Optional


class Watchdog:
    """Arms a countdown around a unit of work; fires if it doesn't finish.

    >>> wd = Watchdog(timeout_s=600)
    >>> with wd.guard("gen 12"):
    ...     engine.step(ranker)   # hangs -> stacks dumped, process exits 124

    The default action dumps all thread stacks to stderr and ``os._exit(124)``
    — a deliberate hard exit: a rank stuck inside a collective cannot run
    Python cleanup, and a fast death lets the launcher restart the job from
    the checkpoint ring instead of holding the node.
    """

    def __init__(self, timeout_s: float, on_timeout: Optional[Callable] = None):
        self.timeout_s = float(timeout_s)
        self.on_timeout = on_timeout or self._default_action
        self._deadline: Optional[float] = None
        self._label = ""
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._fired = False
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def _loop(self):
        while not self._stop.wait(min(1.0, self.timeout_s / 4)):
            with self._lock:
                expired = self._deadline is not None and time.monotonic() > self._deadline
                label = self._label
            if expired and not self._fired:
                self._fired = True
                self.on_timeout(label)

    def _default_action(self, label: str):
        print(f"[watchdog] {label!r} exceeded {self.timeout_s}s — dumping stacks "
              "and exiting", file=sys.stderr, flush=True)
        faulthandler.dump_traceback(file=sys.stderr)
        os._exit(124)

    def arm(self, label: str = ""):
        with self._lock:
            self._label = label
            self._deadline = time.monotonic() + self.timeout_s
            self._fired = False  # a custom on_timeout may not exit; re-arm

    def disarm(self):
        with self._lock:
            self._deadline = None

    def guard(self, label: str = ""):
        wd = self

        class _Guard:
            def __enter__(self):
                wd.arm(label)

            def __exit__(self, *a):
                wd.disarm()

        return _Guard()

    def close(self):
        self._stop.set()
        self._thread.join(timeout=5)


class Heartbeat:
    """Per-rank atomic beat files for external liveness monitoring."""

    def __init__(self, folder: str, rank: int):
        self.folder = folder
        self.rank = rank
        os.makedirs(folder, exist_ok=True)

    def beat(self, gen: int):
        p = os.path.join(self.folder, f"rank{self.rank}.json")
        tmp = p + ".tmp"
        with open(tmp, "w") as f:
            json.dump({"rank": self.rank, "gen": int(gen), "ts": time.time()}, f)
        os.replace(tmp, p)

    @staticmethod
    def read(folder: str) -> List[Dict]:
        out = []
        if not os.path.isdir(folder):
            return out
        for f in sorted(os.listdir(folder)):
            if f.startswith("rank") and f.endswith(".json"):
                try:
                    with open(os.path.join(folder, f)) as fh:
                        out.append(json.load(fh))
                except (OSError, json.JSONDecodeError):
                    pass  # a beat mid-replace; the next read sees it
        return out

    @staticmethod
    def stalled_ranks(folder: str, timeout_s: float) -> List[Dict]:
        """Beats older than ``timeout_s`` — ranks that stopped progressing."""
        now = time.time()
        return [b for b in Heartbeat.read(folder) if now - b["ts"] > timeout_s]


def _main(argv=None):
    import argparse

    p = argparse.ArgumentParser(description="heartbeat liveness check")
    p.add_argument("folder", help="heartbeat dir, e.g. saved/<run>/heartbeat")
    p.add_argument("--timeout", type=float, default=300.0,
                   help="seconds without a beat before a rank counts as stalled")
    args = p.parse_args(argv)
    beats = Heartbeat.read(args.folder)
    stalled = Heartbeat.stalled_ranks(args.folder, args.timeout)
    now = time.time()
    for b in beats:
        mark = "STALLED" if b in stalled else "ok"
        print(f"rank {b['rank']}: gen {b['gen']}, last beat {now - b['ts']:.0f}s "
              f"ago [{mark}]")
    return 1 if stalled else 0


if __name__ == "__main__":
    sys.exit(_main())
