from es_pytorch_amd.nn.nn import BaseNet, FeedForward, FFIntegGausAction, FFIntegGausActionMulti, FFBinned  # noqa: F401
from es_pytorch_amd.nn.optimizers import Optimizer, SimpleES, SGD, Adam  # noqa: F401
from es_pytorch_amd.nn.obstat import ObStat  # noqa: F401
