"""es_pytorch_amd — an MI355X-native Evolution Strategies framework.

A from-scratch re-design of the capabilities of sash-a/es_pytorch
(OpenAI-ES / novelty-search trainer, reference at /root/reference) for AMD
Instinct MI355X (gfx950, CDNA4):

* the node-shared MPI noise window (reference ``src/core/noisetable.py:13-24``)
  becomes a per-GPU HBM-resident table filled by a Philox HIP kernel;
* the sequential per-process gym rollouts (``src/gym/gym_runner.py:33-67``)
  become population-batched on-device rollouts with a fused HIP MLP forward;
* mpi4py collectives (``src/core/es.py:77-91``) become RCCL collectives over
  xGMI via ``torch.distributed`` — one process per GPU;
* the numpy gradient reconstruction (``src/utils/utils.py:29-39``) and Adam
  (``src/nn/optimizers.py:47-61``) become hand-written CDNA4 HIP kernels.

Public API mirrors the reference: ``Policy``, ``NoiseTable``, ``Optimizer``
family, ``Ranker`` family, ``TrainingResult`` family, ``Reporter`` family and
``es.step / es.test_params / es.approx_grad``.
"""

__version__ = "0.1.0"

from es_pytorch_amd.config import load_config, AttrDict  # noqa: F401
