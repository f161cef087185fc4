"""Argument validation/derivation (parity with reference
`peft_pretraining/args_utils.py:8-86`): YAML `--training_config` fully
replaces CLI args; dataset-path XOR megatron-config (XOR synthetic-data,
our offline extension); batch-size algebra; token budget → steps; fp16
forbidden; exactly-one optimizer-reset mode; `--relora` implies
`--use_peft`; `--skip_batches` CSV → set."""

import os
import sys

import yaml

from relora_amd.utils.logging import logger


def check_args_torchrun_main(args, n_cli_args=None):
    if args.training_config is not None:
        logger.info(
            f"Yaml config provided for the run. The file {args.training_config} is used to provide all the parameters."
        )
        if n_cli_args is None:
            n_cli_args = len(sys.argv) - 1
        if n_cli_args > 2:
            logger.error(f"got {n_cli_args} command line arguments")
            raise RuntimeError(
                "You provided both a yaml config and command line arguments. "
                "Please use only one of the two options."
            )
        with open(args.training_config) as f:
            training_config = yaml.safe_load(f)
        for k, v in training_config.items():
            if k == "lr":
                v = float(v)
            setattr(args, k, v)

    n_data_sources = sum(
        x is not None for x in
        (args.dataset_path, args.megatron_dataset_config, getattr(args, "synthetic_data", None))
    )
    if n_data_sources != 1:
        raise ValueError(
            "Exactly one of --dataset_path, --megatron_dataset_config, --synthetic_data "
            f"must be specified. Got {args.dataset_path=}, {args.megatron_dataset_config=}, "
            f"synthetic_data={getattr(args, 'synthetic_data', None)}"
        )

    if args.megatron_dataset_config is not None and not os.path.exists(args.megatron_dataset_config):
        raise ValueError(f"{args.megatron_dataset_config=} does not exist")

    if args.batch_size is None:
        raise ValueError("batch_size must be specified")
    if args.batch_size == "auto" and args.total_batch_size is None:
        raise ValueError("--batch_size auto requires --total_batch_size")

    if args.tags is not None and isinstance(args.tags, str):
        args.tags = args.tags.split(",")

    # NOTE: the reference performs this check after zeroing args.relora for
    # non-peft runs, making it unreachable; we apply the documented intent.
    if args.relora and not args.use_peft:
        logger.warning("--relora assumes --use_peft. Setting --use_peft=True")
        args.use_peft = True

    if not args.use_peft:
        # just for more clear hparam logging
        args.relora = None
        args.lora_r = None
        args.force_keep_original = False

    if args.batch_size != "auto":
        if args.total_batch_size is None:
            args.gradient_accumulation = args.gradient_accumulation or 1
            args.total_batch_size = args.batch_size * args.gradient_accumulation
        assert args.total_batch_size % args.batch_size == 0, \
            "total_batch_size must be divisible by batch_size"

    if args.max_train_tokens is not None:
        args.num_training_steps = args.max_train_tokens // args.total_batch_size
        logger.info(f"Training for {args.num_training_steps} update steps")

    if args.warmed_up_model is not None:
        assert os.path.exists(args.warmed_up_model), f"{args.warmed_up_model=} does not exist"

    if args.dtype in ["fp16", "float16"]:
        raise NotImplementedError("fp16 is not supported; use bfloat16 or float32")

    if (
        int(args.reset_optimizer_on_relora)
        + int(bool(args.optimizer_random_pruning))
        + int(bool(args.optimizer_magnitude_pruning))
    ) > 1:
        raise ValueError(
            "reset_optimizer_on_relora, optimizer_random_pruning and "
            "optimizer_magnitude_pruning are mutually exclusive"
        )

    assert 0 <= args.optimizer_random_pruning < 1, "--optimizer_random_pruning must be between 0 and 1"
    assert 0 <= args.optimizer_magnitude_pruning < 1, "--optimizer_magnitude_pruning must be between 0 and 1"

    if args.distributed_type == "fsdp":
        raise NotImplementedError(
            "FSDP is not supported (it is hard-disabled in the reference too); use ddp"
        )

    if args.skip_batches is not None and isinstance(args.skip_batches, str):
        args.skip_batches = set(map(int, args.skip_batches.split(",")))
        logger.info(f"Skipping batches {args.skip_batches}")
    args.skip_batches = args.skip_batches or set()

    return args
