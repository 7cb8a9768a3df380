from es_pytorch_amd.parallel.comm import Comm, init_comm  # noqa: F401
