from relora_amd.parallel.ddp import DistributedModel  # noqa: F401
from relora_amd.parallel.zero import ZeroRedundancyAdamW  # noqa: F401
