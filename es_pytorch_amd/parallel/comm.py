"""Distribution layer: one process per GPU, RCCL over xGMI.

Replaces the reference's mpi4py usage (call-site inventory in SURVEY.md §2.4):

* fitness-triple exchange — reference implements an allgather as a replicated
  ``comm.Alltoall`` (``src/core/es.py:84-95``); here it is a single
  ``all_gather_into_tensor`` on device buffers (RCCL over xGMI when the
  backend is nccl, gloo for CPU tests);
* ObStat merge — reference uses a custom pickling MPI reduce op
  (``src/nn/obstat.py:5-10,39-43``); here a packed fp64 all_reduce;
* seed / archive / control-state distribution — reference uses
  ``comm.scatter([x]*size)`` broadcast idioms (``src/utils/utils.py:58,69``,
  ``src/utils/novelty.py:10``, ``nsra.py:117-133``); here object broadcasts.

Design contract preserved from the reference (README.md:10-12): parameters are
NEVER communicated — only (fit+, fit-, noise_idx) triples move per generation,
and every rank recomputes the identical ranking/gradient/update redundantly.

With no initialized process group, ``Comm`` degrades to a correct
single-process implementation, so every code path also runs serially.
"""
from __future__ import annotations

import datetime
import os
from typing import Any, List, Optional

import numpy as np
import torch
import torch.distributed as dist


class Comm:
    """Thin collective wrapper with a single-process fallback."""

    def __init__(self, device: Optional[torch.device] = None):
        self.initialized = dist.is_available() and dist.is_initialized()
        if self.initialized:
            self.rank = dist.get_rank()
            self.size = dist.get_world_size()
            self.backend = dist.get_backend()
        else:
            self.rank = 0
            self.size = 1
            self.backend = None
        if device is None:
            if torch.cuda.is_available():
                # modulo device count: N ranks on an M<N-GPU box (twin-rank
                # integration tests) share devices instead of crashing
                local = int(os.environ.get("LOCAL_RANK",
                                           self.rank)) % max(1, torch.cuda.device_count())
                device = torch.device("cuda", local)
            else:
                device = torch.device("cpu")
        self.device = device
        # Collective device: nccl collectives need device tensors; gloo needs CPU.
        self._coll_device = self.device if self.backend == "nccl" else torch.device("cpu")

    # -- collectives ------------------------------------------------------
    def allgather_rows(self, rows: torch.Tensor) -> torch.Tensor:
        """All-gather equal-size 2-D row blocks; returns (size*n, cols) in rank order.

        The RCCL replacement for the reference's replicated Alltoall of
        (fit+, fit-, idx) rows (``src/core/es.py:89-91``).
        """
        if not self.initialized or self.size == 1:
            return rows
        rows = rows.contiguous().to(self._coll_device)
        out = torch.empty((self.size * rows.shape[0], rows.shape[1]), dtype=rows.dtype,
                          device=self._coll_device)
        dist.all_gather_into_tensor(out, rows)
        return out

    def allreduce_sum_(self, t: torch.Tensor) -> torch.Tensor:
        """In-place sum all-reduce (reference ``comm.allreduce(MPI.SUM)``)."""
        if self.initialized and self.size > 1:
            moved = t.device != self._coll_device
            buf = t.to(self._coll_device) if moved else t
            dist.all_reduce(buf, op=dist.ReduceOp.SUM)
            if moved:
                t.copy_(buf.to(t.device))
        return t

    def allreduce_scalar(self, x: float) -> float:
        t = torch.tensor([float(x)], dtype=torch.float64, device=self._coll_device)
        self.allreduce_sum_(t)
        return t.item()

    def broadcast_tensor_(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        """In-place tensor broadcast (device buffers under nccl=RCCL)."""
        if not self.initialized or self.size == 1:
            return t
        moved = t.device != self._coll_device
        buf = t.to(self._coll_device) if moved else t
        dist.broadcast(buf, src=src)
        if moved:
            t.copy_(buf.to(t.device))
        return t

    def broadcast_obj(self, obj: Any, src: int = 0) -> Any:
        """Broadcast a picklable object (reference ``comm.scatter([x]*size)`` idiom)."""
        if not self.initialized or self.size == 1:
            return obj
        box = [obj if self.rank == src else None]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def barrier(self):
        if self.initialized and self.size > 1:
            dist.barrier()

    def allgather_obj(self, obj: Any) -> List[Any]:
        if not self.initialized or self.size == 1:
            return [obj]
        out = [None] * self.size
        dist.all_gather_object(out, obj)
        return out


def init_comm(device: Optional[torch.device] = None, timeout_s: int = 600) -> Comm:
    """Initialize torch.distributed from torchrun env vars if present.

    One process per GPU; backend nccl (=RCCL on ROCm) when CUDA devices are
    visible, gloo otherwise. No-op (single-process Comm) without WORLD_SIZE.
    """
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws > 1 and not (dist.is_available() and dist.is_initialized()):
        # ES_COMM_BACKEND=gloo forces gloo transport while compute stays on
        # the GPU — lets N ranks share ONE physical GPU for integration tests
        # (RCCL, like NCCL, rejects two ranks on the same device)
        backend = os.environ.get(
            "ES_COMM_BACKEND", "nccl" if torch.cuda.is_available() else "gloo")
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", "0"))
            torch.cuda.set_device(local % max(1, torch.cuda.device_count()))
        dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    return Comm(device)


def seed_all(comm: Comm, seed, n_rng_streams: int = 1):
    """Seeding semantics of reference ``src/utils/utils.py:61-76``:

    * every rank gets a DISTINCT numpy RandomState (for noise-index draws),
    * torch is seeded IDENTICALLY on every rank (rank-0's seed broadcast) so
      initial network params are replicated without a parameter broadcast.

    :returns: (per-rank RandomState, my_seed, global torch seed)
    """
    if seed is not None and hasattr(seed, "__len__") and len(seed) == comm.size:
        my_seed = int(seed[comm.rank])
    elif seed is not None and not hasattr(seed, "__len__"):
        my_seed = int(seed) + comm.rank
    else:
        my_seed = int.from_bytes(os.urandom(4), "little")
    rs = np.random.RandomState(my_seed)

    global_seed = comm.broadcast_obj(my_seed, src=0)
    torch.random.manual_seed(global_seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(global_seed)
    return rs, my_seed, global_seed
