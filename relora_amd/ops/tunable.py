"""hipBLASLt/rocBLAS GEMM algorithm selection via PyTorch TunableOp.

A pre-tuned table for MI355X (gfx950) ships in
relora_amd/tuning/tunableop_mi355x.csv (measured +3.7% end-to-end on the
llama_1b flagship step vs the default heuristic pick).  `enable_tuned_gemms`
loads it in replay-only mode: no runtime tuning cost, silently skipped on
other architectures or when TunableOp is unavailable.

Re-tune (on an MI355X box) with:
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=relora_amd/tuning/tunableop_mi355x.csv \
      python bench.py --steps 2 --warmup 1
"""

import os

import torch

from relora_amd.utils.logging import logger

_TABLE = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "tuning", "tunableop_mi355x.csv")
_enabled = False


def enable_tuned_gemms():
    """Idempotent; call before the first GEMM. Replay-only (no tuning)."""
    global _enabled
    if _enabled or not torch.cuda.is_available():
        return False
    if os.environ.get("RELORA_AMD_TUNABLEOP", "1") == "0":
        return False
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") == "1":
        # an explicit tuning run manages TunableOp itself
        return False
    try:
        if "gfx950" not in torch.cuda.get_device_properties(0).gcnArchName:
            return False
        t = torch.cuda.tunable
        t.enable(True)
        t.tuning_enable(False)
        if os.path.exists(_TABLE):
            t.set_filename(_TABLE)
            t.read_file(_TABLE)
        _enabled = True
        logger.info(f"TunableOp GEMM table loaded from {_TABLE}")
        return True
    except Exception as e:  # pragma: no cover
        logger.warning(f"TunableOp unavailable: {e}")
        return False
