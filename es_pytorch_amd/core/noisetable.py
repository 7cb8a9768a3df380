"""Per-GPU HBM-resident noise table.

Reference design (``src/core/noisetable.py``): ONE 1 GB float32 block of
N(0,1) per node in an MPI shared-memory window, filled by one rank per node
from a seed handed out over send/recv, with a global barrier (SURVEY.md
C4-C8). A perturbation is a random scalar offset into the table; the slice of
length n_params is a zero-copy view.

MI355X-native design: no host window and no fill traffic at all — the seed is
broadcast once (RCCL) and EVERY GPU fills its own HBM-resident
replica deterministically with the Philox kernel (``ops/csrc/hip/noise.hip``),
in milliseconds. 1-8 GB in 288 GB HBM is trivial; multi-GB tables are a
config knob, not an architecture change. On CPU the same Philox fill runs via
the g++-compiled twin (``ops/csrc/cpu_ops.cpp``) so tests and the episodic
path see a consistent table.

API mirrors the reference class: ``get/sample_idx/sample/__getitem__/__len__``
(``noisetable.py:33-58``) with torch tensors instead of numpy views.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from es_pytorch_amd import ops
from es_pytorch_amd.parallel.comm import Comm


class NoiseTable:
    def __init__(self, n_params: int, noise: torch.Tensor):
        assert noise.dtype == torch.float32 and noise.dim() == 1
        self.n_params = int(n_params)
        self.noise = noise
        self._size = noise.numel()

    def get(self, i: int, size: int) -> torch.Tensor:
        assert len(self) > i + size, "trying to index outside the range of the noise table"
        return self.noise[i:i + size]

    def sample_idx(self, rs: np.random.RandomState, size: int) -> int:
        upper_bound = len(self) - size
        if upper_bound <= 0:
            raise ValueError(f"Network (size:{size}) is too large for noise table (size:{len(self)})")
        return int(rs.randint(0, upper_bound))

    def sample_idxs(self, rs: np.random.RandomState, n: int, size: Optional[int] = None) -> np.ndarray:
        """Batched index draw for the population engine (one call per generation)."""
        size = self.n_params if size is None else size
        upper_bound = len(self) - size
        if upper_bound <= 0:
            raise ValueError(f"Network (size:{size}) is too large for noise table (size:{len(self)})")
        return rs.randint(0, upper_bound, size=n).astype(np.int64)

    def sample(self, rs: Optional[np.random.RandomState] = None, size: Optional[int] = None) \
            -> Tuple[int, torch.Tensor]:
        if size is None:
            size = self.n_params
        if rs is None:
            rs = np.random.RandomState()
        idx = self.sample_idx(rs, size)
        return idx, self.get(idx, size)

    def __getitem__(self, item) -> torch.Tensor:
        return self.get(item, self.n_params)

    def __len__(self):
        return self._size

    def __call__(self, *args, **kwargs) -> Tuple[int, torch.Tensor]:
        return self.sample()

    @staticmethod
    def make_noise(size: int, seed: int, device: Optional[torch.device] = None,
                   stream_id: int = 0) -> torch.Tensor:
        """Deterministic N(0,1) fill via the shared Philox implementation.

        Same (seed, index)->value mapping on CPU and GPU (up to libm/ocml ULP
        in the Box-Muller transcendentals; parity-tested on device).
        """
        device = torch.device("cpu") if device is None else torch.device(device)
        t = torch.empty(size, dtype=torch.float32, device=device)
        if device.type == "cuda":
            with torch.cuda.device(device):
                stream = torch.cuda.current_stream(device).cuda_stream
                ops.check(ops.hip().es_noise_fill(t.data_ptr(), size, seed & (2**64 - 1),
                                                  stream_id, stream), "es_noise_fill")
        else:
            ops.cpu().es_noise_fill_cpu(t.data_ptr(), size, seed & (2**64 - 1), stream_id)
        return t

    @staticmethod
    def create_shared(comm: Comm, size: int, n_params: int, reporter=None,
                      seed: Optional[int] = None, device: Optional[torch.device] = None) -> "NoiseTable":
        """Build the per-rank noise table replica.

        Replaces the reference's shared-window choreography
        (``noisetable.py:66-91``: Split_type + Win.Allocate_shared + seed
        send/recv + rank-1 fill + Barrier) with: rank-0 draws/uses the seed,
        ONE broadcast (SURVEY.md C7 -> RCCL broadcast), every rank fills its
        own device table deterministically, one barrier.
        """
        device = device if device is not None else comm.device
        if comm.rank == 0:
            if seed is None:
                seed = int(np.random.randint(0, 1_000_000))
            if reporter is not None:
                reporter.print(f"nt seed:{seed}")
        seed = comm.broadcast_obj(seed, src=0)
        noise = NoiseTable.make_noise(size, seed, device)
        comm.barrier()
        return NoiseTable(n_params, noise)
