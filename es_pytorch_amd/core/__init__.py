from es_pytorch_amd.core.noisetable import NoiseTable  # noqa: F401
from es_pytorch_amd.core.policy import Policy  # noqa: F401
