"""ES generation step — the algorithm core.

Same control flow and public signatures as the reference
(``src/core/es.py:23-101``): one generation = sample antithetic perturbations,
evaluate via the injected ``fit_fn`` (inversion of control — the core never
imports an environment), share (fit+, fit-, noise_idx) triples,
rank, reconstruct the gradient as a fitness-weighted noise sum, optimizer
step, noiseless eval, report.

Communication: the reference's replicated ``comm.Alltoall``
(``es.py:84-95``) becomes ONE RCCL all_gather of an fp64 (E, 2*O+1) row block
(``parallel/comm.py``); the steps counter all_reduce (``es.py:79``) and the
ObStat merge (``obstat.py:39-43``) become RCCL all_reduces. Parameters are
never communicated; every rank recomputes the identical ranking + gradient +
update (reference README.md:10-12).

This module is the EPISODIC path (works on CPU, any env, any fit_fn). The
GPU-batched whole-generation path with the same semantics lives in
``core/engine.py``.


PROVENANCE: the step/test_params/approx_grad signatures and the generation
control flow are deliberately ported from the reference (src/core/es.py) —
they ARE the public API contract this framework preserves; the collective
layer underneath (parallel/comm.py) and the GPU whole-generation engine
(core/engine.py) are original MI355X-native code.
"""
from __future__ import annotations

from typing import Callable, List, Tuple

import numpy as np
import torch

from es_pytorch_amd.core.noisetable import NoiseTable
from es_pytorch_amd.core.policy import Policy
from es_pytorch_amd.nn.obstat import ObStat
from es_pytorch_amd.parallel.comm import Comm
from es_pytorch_amd.rollout.results import TrainingResult
from es_pytorch_amd.utils.rankers import Ranker
from es_pytorch_amd.utils.utils import scale_noise


def step(cfg, comm: Comm, policy: Policy, nt: NoiseTable, env,
         fit_fn: Callable[[torch.nn.Module], TrainingResult],
         rs: np.random.RandomState, ranker: Ranker, reporter) -> Tuple[TrainingResult, ObStat]:
    """Run a single generation of ES (reference ``es.py:23-51``).

    :param fit_fn: evaluates a phenotype, returns a :class:`TrainingResult`
    :returns: (noiseless TrainingResult, this generation's ObStat)
    """
    assert cfg.general.policies_per_gen % comm.size == 0 and \
        (cfg.general.policies_per_gen / comm.size) % 2 == 0
    eps_per_proc = int((cfg.general.policies_per_gen / comm.size) / 2)

    ob_shape = env.observation_space.shape if env is not None else policy._module._obmean.shape
    gen_obstat = ObStat(ob_shape, 0)
    pos_res, neg_res, inds, steps = test_params(comm, eps_per_proc, policy, nt, gen_obstat,
                                                fit_fn, rs)

    reporter.print(f"n dupes: {len(inds) - len(set(inds.tolist()))}")

    ranker.rank(pos_res, neg_res, inds)
    approx_grad(policy, ranker, nt, policy.flat_params, cfg.general.batch_size,
                cfg.policy.l2coeff)
    noiseless_result = fit_fn(policy.pheno(np.zeros(len(policy), dtype=np.float32)), False)
    reporter.log_gen(ranker.fits, noiseless_result, policy, steps)

    return noiseless_result, gen_obstat


def test_params(comm: Comm, n: int, policy: Policy, nt: NoiseTable, gen_obstat: ObStat,
                fit_fn: Callable[[torch.nn.Module], TrainingResult],
                rs: np.random.RandomState) -> Tuple[np.ndarray, np.ndarray, np.ndarray, int]:
    """Evaluate ``n`` antithetic perturbation pairs on this rank and share
    results with all ranks (reference ``es.py:54-81``).

    positive_results[i] is the fitness with nt[noise_inds[i]] ADDED to the
    params, negative_results[i] with the same noise subtracted.

    :returns: (all positive results, all negative results, all noise inds, total steps)
    """
    results_pos: List[TrainingResult] = []
    results_neg: List[TrainingResult] = []
    inds: List[int] = []
    for _ in range(n):
        idx, noise = nt.sample(rs)
        inds.append(idx)
        results_pos.append(fit_fn(policy.pheno(noise)))
        results_neg.append(fit_fn(policy.pheno(-noise)))
        gen_obstat.inc(*results_pos[-1].ob_sum_sq_cnt)
        gen_obstat.inc(*results_neg[-1].ob_sum_sq_cnt)

    n_objectives = len(results_pos[0].result)
    results = _share_results(comm, [tr.result for tr in results_pos],
                             [tr.result for tr in results_neg], inds)
    gen_obstat.dist_inc(comm)
    steps = int(comm.allreduce_scalar(sum(tr.steps for tr in results_pos + results_neg)))

    return (results[:, 0:n_objectives], results[:, n_objectives:2 * n_objectives],
            results[:, -1], steps)


def _share_results(comm: Comm, fits_pos: List[List[float]], fits_neg: List[List[float]],
                   inds: List[int]) -> np.ndarray:
    """Share fitness triples to all ranks (reference ``es.py:84-95``).

    The reference replicates each rank's rows world-size times and Alltoalls;
    here it is one all_gather of the (E, 2*O+1) fp64 row block — same result
    (rows in rank order), a fraction of the traffic.
    """
    rows = np.array([fp + fn + [i] for fp, fn, i in zip(fits_pos, fits_neg, inds)],
                    dtype=np.float64)
    gathered = comm.allgather_rows(torch.from_numpy(rows))
    return gathered.cpu().numpy().reshape((-1, rows.shape[1]))


def approx_grad(policy: Policy, ranker: Ranker, nt: NoiseTable, params: np.ndarray,
                batch_size: int, l2coeff: float):
    """Approximate the gradient and update policy params (reference ``es.py:98-101``)."""
    grad = scale_noise(ranker.ranked_fits, ranker.noise_inds, nt, len(policy), batch_size)
    grad = (grad / ranker.n_fits_ranked).cpu().numpy()
    policy.optim_step(l2coeff * params - grad)
