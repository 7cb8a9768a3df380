"""RCCL library smoke (world_size=1).

RCCL rejects two ranks on one device, so a 1-GPU box cannot run a real
multi-rank collective — but it CAN initialize the nccl(=RCCL) backend and
run degenerate collectives through it, which exercises the library load,
communicator setup and kernel launches of the exact transport the driver's
multi-GPU run uses. Failures here (missing librccl, HSA IPC config, comm
init hangs) would otherwise only surface mid-scale-run."""
import datetime
import socket

import numpy as np
import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available(), reason="GPU only")
def test_rccl_initializes_and_reduces():
    if dist.is_initialized():
        pytest.skip("a process group already exists in this process")
    with socket.socket() as sock:  # dynamically allocated free port
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
    store = dist.TCPStore("127.0.0.1", port, 1, True,
                          datetime.timedelta(seconds=60))
    dist.init_process_group("nccl", store=store, rank=0, world_size=1,
                            timeout=datetime.timedelta(seconds=60))
    try:
        assert dist.get_backend() == "nccl"
        t = torch.arange(8, dtype=torch.float64, device="cuda:0")
        dist.all_reduce(t)  # degenerate (world 1) but runs through RCCL
        np.testing.assert_array_equal(t.cpu().numpy(), np.arange(8.0))
        dist.broadcast(t, src=0)
        out = torch.empty(8, dtype=torch.float64, device="cuda:0")
        dist.all_gather_into_tensor(out, t)
        np.testing.assert_array_equal(out.cpu().numpy(), np.arange(8.0))
        dist.barrier()
    finally:
        dist.destroy_process_group()
