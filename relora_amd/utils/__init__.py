from relora_amd.utils.logging import logger  # noqa: F401
from relora_amd.utils.wandb_shim import wandb  # noqa: F401
