"""Distributed ReLoRA pretraining entry point (CLI parity with the
reference `torchrun_main.py`).

Usage (one process per GPU, RCCL over xGMI):
    torchrun --nproc-per-node 8 torchrun_main.py --model_config configs/llama_1b.json \
        --use_peft true --relora 5000 --cycle_length 5000 ...
or with a yaml recipe:
    torchrun --nproc-per-node 8 torchrun_main.py --training_config training_configs/1B_v1.0.yaml
"""

import os

from relora_amd.trainer import main, parse_args

if __name__ == "__main__":
    if os.environ.get("RELORA_AMD_HANG_DUMP_S"):
        import faulthandler
        faulthandler.dump_traceback_later(
            int(os.environ["RELORA_AMD_HANG_DUMP_S"]), repeat=True)
    args = parse_args()
    main(args)
