"""Run-level checkpointing: an atomic ring of full-run snapshots + exact resume.

Beyond reference parity: the reference checkpoints only the Policy pickle
(``src/utils/reporters.py:177-188``, resumed via ``obj.py:39-41``), so a
resumed run repeats neither schedules nor RNG streams. A RunCheckpointer
snapshot carries EVERYTHING the next generation depends on — the Policy
pickle bytes (params, ObStat, optimizer state; reference format), every
rank's numpy RandomState and torch RNG state, the mutated config scalars
(lr/std/ac_std decay schedules), the generation index and arbitrary extra
state (novelty archive, stagnation counters) — so a killed run resumes
bit-for-bit: N generations straight equals k generations + kill + resume +
N-k generations (``tests/test_checkpoint.py``).

Write protocol (rank 0): gather per-rank RNG states over the collective,
pickle to ``ckpt-<gen>.pkl.tmp``, fsync, ``os.replace`` into place — a crash
mid-write can never corrupt the newest good snapshot — then prune the ring
to the ``keep`` newest. All ranks barrier after the write so no rank runs
ahead of durable state.
"""
from __future__ import annotations

import os
import pickle
import random
import re
from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.parallel.comm import Comm

_CKPT_RE = re.compile(r"^ckpt-(\d+)\.pkl$")


class RunCheckpointer:
    def __init__(self, folder: str, comm: Comm, keep: int = 3, every: int = 1):
        self.folder = folder
        self.comm = comm
        self.keep = max(1, int(keep))
        self.every = max(1, int(every))

    # ----------------------------------------------------------------- save
    def maybe_save(self, next_gen: int, policy: Policy, rs: np.random.RandomState,
                   cfg=None, engine=None, env=None,
                   extra: Optional[Dict[str, Any]] = None) -> bool:
        """Save iff ``next_gen`` lands on the ``every`` cadence."""
        if next_gen % self.every != 0:
            return False
        self.save(next_gen, policy, rs, cfg=cfg, engine=engine, env=env, extra=extra)
        return True

    def save(self, next_gen: int, policy: Policy, rs: np.random.RandomState,
             cfg=None, engine=None, env=None,
             extra: Optional[Dict[str, Any]] = None) -> str:
        """Snapshot state such that the run continues from generation
        ``next_gen`` exactly as if it had never stopped."""
        engines = _as_list(engine)
        for e in engines:
            e.sync_host(light=False)  # device truth -> Policy incl. moments
        # identical updates are recomputed on every rank (reference es.py:98-101
        # design), so rank 0's policy is THE policy; RNG streams are per-rank
        # (numpy rs, torch, and python random — nsra.py's policy selection)
        rng_states = self.comm.allgather_obj((rs.get_state(),
                                              torch.get_rng_state().numpy(),
                                              random.getstate()))
        # per-rank engine internals (gen counter + prefetched offset draws)
        eng_states = self.comm.allgather_obj(
            [e.checkpoint_state() for e in engines] or None)
        path = os.path.join(self.folder, f"ckpt-{next_gen}.pkl")
        if self.comm.rank == 0:
            os.makedirs(self.folder, exist_ok=True)
            state = {
                "next_gen": int(next_gen),
                "world_size": self.comm.size,
                "policy": pickle.dumps(policy),
                "rng_states": rng_states,
                "cfg": _plain(cfg) if cfg is not None else None,
                "engine_states": eng_states,
                # episodic CPU envs count episodes via a per-env seed counter
                # (envs/base.py SingleFromBatched._seed); GPU envs are
                # reseeded from engine.gen each generation and need nothing
                "env_seed": getattr(env, "_seed", None),
                "extra": extra or {},
            }
            tmp = path + ".tmp"
            with open(tmp, "wb") as f:
                pickle.dump(state, f)
                f.flush()
                os.fsync(f.fileno())
            os.replace(tmp, path)
            self._prune()
        self.comm.barrier()
        return path

    def _prune(self):
        snaps = self._snapshots()
        for _, p in snaps[:-self.keep]:
            try:
                os.unlink(p)
            except OSError:
                pass

    def _snapshots(self):
        if not os.path.isdir(self.folder):
            return []
        out = []
        for f in os.listdir(self.folder):
            m = _CKPT_RE.match(f)
            if m:
                out.append((int(m.group(1)), os.path.join(self.folder, f)))
        return sorted(out)

    # -------------------------------------------------------------- restore
    def latest(self) -> Optional[str]:
        snaps = self._snapshots()
        return snaps[-1][1] if snaps else None

    def load(self, path: Optional[str] = None) -> Optional[Dict[str, Any]]:
        path = path or self.latest()
        if path is None:
            return None
        with open(path, "rb") as f:
            return pickle.load(f)

    def restore(self, state: Dict[str, Any], policy: Policy,
                rs: np.random.RandomState, cfg=None, engine=None,
                env=None, allow_reshard: bool = False) -> Tuple[int, Dict[str, Any]]:
        """Restore IN PLACE into the live objects; returns (next_gen, extra).

        With ``allow_reshard=True`` a snapshot written at a different world
        size restores too (elastic restart, e.g. an 8-GPU run resumed on 4):
        learned state — params, moments, ObStat, schedules, archive — carries
        over exactly, but the per-rank RNG streams must re-split, so the
        continuation is deterministic given (snapshot, new world size) yet
        not bitwise-identical to the uninterrupted original. Default strict
        mode refuses the mismatch so same-size resumes stay bit-exact.
        """
        resharded = state["world_size"] != self.comm.size
        if resharded and not allow_reshard:
            raise RuntimeError(
                f"checkpoint was written by world_size={state['world_size']}, "
                f"resuming with {self.comm.size} ranks would desync the RNG "
                "streams — relaunch with the original rank count, or pass "
                "allow_reshard=True for a deterministic (not bitwise) restart")
        saved = pickle.loads(state["policy"])
        for live, snap in zip(_as_list(policy), _as_list(saved)):
            _restore_policy(live, snap)
        if resharded:
            # fresh deterministic per-rank streams for the new rank layout
            mix = (int(state["next_gen"]) * 1000003 + self.comm.rank * 7919
                   + self.comm.size * 104729) & 0x7FFFFFFF
            rs.set_state(np.random.RandomState(mix).get_state())
            torch.manual_seed(int(state["next_gen"]))
            random.seed(mix)
        else:
            np_state, torch_state, py_state = state["rng_states"][self.comm.rank]
            rs.set_state(np_state)
            torch.set_rng_state(torch.from_numpy(torch_state))
            random.setstate(py_state)
        if cfg is not None and state["cfg"] is not None:
            # restore ONLY the schedule-mutated scalars: run-duration and
            # launch intent (gens, ckpt cadence, env, ...) must come from the
            # NEW invocation, or a relaunch with a larger gens budget would
            # be clobbered back to the finished one and exit immediately
            for sect, key in (("noise", "std"), ("policy", "lr"),
                              ("policy", "ac_std")):
                sv = state["cfg"].get(sect, {})
                if key in sv and sect in cfg and key in cfg[sect]:
                    cfg[sect][key] = sv[key]
        if resharded:
            # prefetched offset draws belong to the OLD rank split; drop them
            for e in _as_list(engine):
                e.restore_from_policy(gen=state["next_gen"])
        else:
            ests = state.get("engine_states",
                             [None] * self.comm.size)[self.comm.rank]
            engines = _as_list(engine)
            if not ests:  # snapshot written without engines (CPU path)
                ests = [None] * len(engines)
            for e, est in zip(engines, ests):
                e.restore_from_policy(gen=state["next_gen"])
                if est is not None:
                    e.load_checkpoint_state(est)
        if env is not None and state.get("env_seed") is not None:
            env._seed = state["env_seed"]
        return state["next_gen"], dict(state["extra"])


def _as_list(x):
    if x is None:
        return []
    return list(x) if isinstance(x, (list, tuple)) else [x]


def _restore_policy(policy: Policy, saved: Policy):
    policy.flat_params[:] = saved.flat_params
    policy.std = saved.std
    policy.optim.__dict__.update(saved.optim.__dict__)
    policy.obstat = saved.obstat
    if hasattr(saved._module, "_action_std"):
        policy._module._action_std = saved._module._action_std
    policy.set_nn_params(policy.flat_params)


def _plain(obj):
    """AttrDict/dict tree -> plain-dict tree (stable pickles)."""
    if isinstance(obj, dict):
        return {k: _plain(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_plain(v) for v in obj)
    return obj
