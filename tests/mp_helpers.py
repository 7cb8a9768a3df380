"""Multi-process test harness: N ranks over gloo on 127.0.0.1.

The reference runs its distributed tests under ``mpirun -np N pytest``
(reference ``test/__init__.py:4-7``); here each test spawns its own ranks so
plain single-process pytest exercises world_size>1 — same assertion pattern:
rank-dependent inputs, rank-independent expected outputs, asserted on every
rank.
"""
from __future__ import annotations

import multiprocessing as mp
import os
import pickle
import socket
import tempfile
import traceback


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank: int, world: int, port: int, fn, args, result_dir: str):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["LOCAL_RANK"] = str(rank)
        import torch.distributed as dist
        dist.init_process_group(backend="gloo", rank=rank, world_size=world)
        out = fn(rank, world, *args)
        with open(os.path.join(result_dir, f"rank{rank}.ok"), "wb") as f:
            pickle.dump(out, f)
        dist.destroy_process_group()
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.err"), "w") as f:
            f.write(traceback.format_exc())
        raise


def run_mp(fn, world: int = 2, args: tuple = (), timeout: int = 120):
    """Run ``fn(rank, world, *args)`` on ``world`` gloo ranks; returns rank results."""
    ctx = mp.get_context("spawn")
    port = _free_port()
    with tempfile.TemporaryDirectory() as d:
        procs = [ctx.Process(target=_entry, args=(r, world, port, fn, args, d))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout)
        errs = []
        for r in range(world):
            ep = os.path.join(d, f"rank{r}.err")
            if os.path.exists(ep):
                errs.append(f"--- rank {r} ---\n" + open(ep).read())
        for p in procs:
            if p.is_alive():
                p.terminate()
                errs.append(f"rank timed out")
        if errs:
            raise AssertionError("\n".join(errs))
        out = []
        for r in range(world):
            with open(os.path.join(d, f"rank{r}.ok"), "rb") as f:
                out.append(pickle.load(f))
        return out
