"""Gradient-reconstruction helpers + misc utilities.

``scale_noise`` is the reference's compute hotspot of the update
(``src/utils/utils.py:29-39``): sum_i fit_i * nt[ind_i : ind_i + n_params].
Here it dispatches to the native gather-GEMV — the HIP kernel
(``ops/csrc/hip/grad.hip``) when the table lives on GPU, the C++ twin on CPU —
with a pure-torch batched fallback that reproduces the reference's batching
(``batch_noise``, ``utils.py:14-26``) for arbitrary tensors.
"""
from __future__ import annotations

from typing import Iterator

import numpy as np
import torch

from es_pytorch_amd import ops
from es_pytorch_amd.core.noisetable import NoiseTable


def batch_noise(inds: np.ndarray, nt: NoiseTable, policy_len: int, batch_size: int) \
        -> Iterator[torch.Tensor]:
    """Yield (batch, n_params) stacks of noise rows (reference ``utils.py:14-26``)."""
    assert inds.ndim == 1
    batch = []
    for idx in inds:
        batch.append(nt.get(int(idx), policy_len))
        if len(batch) == batch_size:
            yield torch.stack(batch)
            batch = []
    if batch:
        yield torch.stack(batch)


def scale_noise(fits: np.ndarray, noise_inds: np.ndarray, nt: NoiseTable, policy_len: int,
                batch_size: int) -> torch.Tensor:
    """Fitness-weighted sum of noise rows (reference ``utils.py:29-39``).

    Native fast path: the gather-GEMV kernel reads each row's slice straight
    from the table (no row copies). Fallback: the reference's batched-dot.
    """
    assert len(fits) == len(noise_inds)
    fits = np.asarray(fits, dtype=np.float32)
    offs = np.asarray(noise_inds, dtype=np.int64)
    dev = nt.noise.device

    if dev.type == "cuda":
        fits_t = torch.from_numpy(fits).to(dev)
        offs_t = torch.from_numpy(offs).to(dev)
        g = torch.empty(policy_len, dtype=torch.float32, device=dev)
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.check(ops.hip().es_grad_gather(g.data_ptr(), nt.noise.data_ptr(),
                                           fits_t.data_ptr(), offs_t.data_ptr(),
                                           len(fits), policy_len, 0.0, stream),
                  "es_grad_gather")
        return g

    try:
        lib = ops.cpu()
        g = torch.empty(policy_len, dtype=torch.float32)
        table = nt.noise
        assert table.is_contiguous()
        ft = torch.from_numpy(fits).contiguous()
        ot = torch.from_numpy(offs).contiguous()
        lib.es_grad_gather_cpu(g.data_ptr(), table.data_ptr(), ft.data_ptr(), ot.data_ptr(),
                               len(fits), policy_len)
        return g
    except Exception:
        # reference-style batched dot fallback
        total = torch.zeros(policy_len, dtype=torch.float32)
        batched_fits = [fits[i:i + batch_size] for i in range(0, len(fits), batch_size)]
        for fit_batch, noise_batch in zip(batched_fits,
                                          batch_noise(offs, nt, policy_len, batch_size)):
            total += torch.from_numpy(fit_batch) @ noise_batch
        return total


def generate_seed(comm) -> int:
    """Rank-0 seed broadcast (reference ``utils.py:56-58``)."""
    return comm.broadcast_obj(int(np.random.randint(0, 1_000_000)), src=0)
