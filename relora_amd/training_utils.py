"""LR schedulers, optimizer-state reset, and checkpoint helpers.

Behavioral parity with the reference `peft_pretraining/training_utils.py`:
`get_scheculer` (sic, :56-100 — name kept for API parity), the
cyclical-cosine and cosine-with-restarts lambdas (:173-236, including the
1e-7 resume guard and the `adjust_step` restart-sync semantics),
`random_pruning_`/`magnitude_pruning_` (:150-170), `optimizer_reset`
(:267-364, incl. the ZeRO `optimizer.optim.state` path and the 0.999
random-prune standing in for a hard zero), `get_last_training_state`
(:248-264), `delete_old_checkpoints` (:406-418),
`max_train_tokens_to_number` (:239-245).

Pruning runs as torch ops (quantile + masked multiply) on whatever device
the optimizer state lives on; it fires once per ReLoRA cycle (every few
thousand steps) and is far off the hot path, so it has no dedicated HIP
kernel.
"""

import json
import math
import os
import shutil
from functools import partial

import torch
import torch.distributed as dist
from torch.optim.lr_scheduler import LambdaLR

from relora_amd.utils.logging import logger
from relora_amd.utils.wandb_shim import wandb


# ---------------------------------------------------------------------------
# schedulers
# ---------------------------------------------------------------------------


def get_scheculer(
    optimizer,
    *,
    scheduler_type,
    num_training_steps,
    warmup_steps,
    min_lr_ratio,
    cycle_length=None,
    restart_warmup_steps=None,
    adjust_step=0,
    last_epoch=-1,
):
    if adjust_step != 0 and scheduler_type != "cosine_restarts":
        raise ValueError("adjust_step is only supported for cosine_restarts scheduler")

    if scheduler_type == "linear":
        lr_lambda = partial(
            _linear_schedule_lambda,
            num_warmup_steps=warmup_steps,
            num_training_steps=num_training_steps,
        )
        return LambdaLR(optimizer, lr_lambda, last_epoch)
    if scheduler_type == "cosine":
        return get_cyclical_cosine_schedule_with_min_lr(
            optimizer,
            num_warmup_steps=warmup_steps,
            num_training_steps=num_training_steps,
            cycle_length=cycle_length,
            min_lr_ratio=min_lr_ratio,
            last_epoch=last_epoch,
        )
    if scheduler_type == "cosine_restarts":
        assert restart_warmup_steps is not None, "restart_warmup_steps must be specified for cosine_restarts scheduler"
        return get_cosine_schedule_with_multiple_warmups(
            optimizer,
            num_training_steps=num_training_steps,
            first_warmup_steps=warmup_steps,
            restart_warmup_steps=restart_warmup_steps,
            restart_every=cycle_length,
            min_lr_ratio=min_lr_ratio,
            adjust_step=adjust_step,
            last_epoch=last_epoch,
        )
    raise NotImplementedError(f"Scheduler {scheduler_type} is not implemented")


# alias with the spelling fixed
get_scheduler = get_scheculer


def _linear_schedule_lambda(current_step, *, num_warmup_steps, num_training_steps):
    if current_step < num_warmup_steps:
        return float(current_step) / float(max(1, num_warmup_steps))
    return max(
        0.0,
        float(num_training_steps - current_step)
        / float(max(1, num_training_steps - num_warmup_steps)),
    )


def get_cyclical_cosine_schedule_with_min_lr(
    optimizer, num_warmup_steps, num_training_steps, cycle_length, min_lr_ratio=0.1, last_epoch=-1
):
    assert cycle_length is not None or num_training_steps is not None, \
        "You must specify either cycle_length or num_training_steps"
    if cycle_length is None:
        cycle_length = num_training_steps
    if num_training_steps % cycle_length != 0:
        raise ValueError(
            f"num_training_steps ({num_training_steps}) must be divisible by cycle_length ({cycle_length})"
        )
    lr_lambda = partial(
        _cyclical_cosine_lambda,
        num_warmup_steps=num_warmup_steps,
        cycle_length=cycle_length,
        min_lr_ratio=min_lr_ratio,
    )
    return LambdaLR(optimizer, lr_lambda, last_epoch)


def _cyclical_cosine_lambda(current_step, *, num_warmup_steps, cycle_length, min_lr_ratio):
    assert 0 < min_lr_ratio <= 1.0, "min_lr_ratio must be in (0,1]"
    cycle_step = current_step % cycle_length

    if cycle_step < num_warmup_steps:
        # resume guard: when replaying a later cycle, don't re-do a full
        # warmup from zero (reference training_utils.py:179-183)
        if current_step != cycle_step and cycle_step < 2:
            return 1e-7
        return float(cycle_step) / float(max(1, num_warmup_steps))

    progress = float(cycle_step - num_warmup_steps) / float(max(1, cycle_length - num_warmup_steps))
    cosine_decay = 0.5 * (1.0 + math.cos(math.pi * progress))
    return min_lr_ratio + (1.0 - min_lr_ratio) * cosine_decay


def get_cosine_schedule_with_multiple_warmups(
    optimizer,
    *,
    num_training_steps,
    first_warmup_steps,
    restart_warmup_steps,
    restart_every,
    min_lr_ratio=0.1,
    adjust_step=0,
    last_epoch=-1,
):
    if restart_every is None:
        raise ValueError("restart_every must be specified for cosine_restarts scheduler")
    if num_training_steps % restart_every != 0:
        raise ValueError(
            f"num_training_steps ({num_training_steps}) must be divisible by restart_every ({restart_every})"
        )
    lr_lambda = partial(
        _cosine_restarts_lambda,
        num_training_steps=num_training_steps,
        first_warmup_steps=first_warmup_steps,
        restart_warmup_steps=restart_warmup_steps,
        restart_every=restart_every,
        min_lr_ratio=min_lr_ratio,
        adjust_step=adjust_step,
    )
    return LambdaLR(optimizer, lr_lambda, last_epoch)


def _cosine_restarts_lambda(
    current_step,
    *,
    num_training_steps,
    first_warmup_steps,
    restart_warmup_steps,
    restart_every,
    min_lr_ratio,
    adjust_step,
):
    """Cosine decay with a short re-warmup after every restart; the re-warmup
    peak tracks the decayed cosine envelope. `adjust_step` shifts the restart
    grid to sync resets after a warm start (reference training_utils.py:191-236)."""
    assert 0 < min_lr_ratio <= 1.0, "min_lr_ratio must be in (0,1]"
    assert restart_every > 0, "restart_every must be positive"
    assert adjust_step + first_warmup_steps <= num_training_steps, \
        "warmup + adjust_step is more than full training steps"
    assert adjust_step + first_warmup_steps <= restart_every, \
        "the first reset will happen before the warmup is done"

    if current_step < first_warmup_steps:
        return float(current_step) / float(max(1, first_warmup_steps))

    _current_step = current_step + adjust_step
    restart_step = _current_step % restart_every
    restart_number = _current_step // restart_every

    if restart_step < restart_warmup_steps and current_step >= restart_every:
        # lr multiplier the envelope will have at the end of this re-warmup
        end_of_warmup_progress = float(
            restart_number * restart_every + restart_warmup_steps - first_warmup_steps
        ) / float(max(1, num_training_steps - first_warmup_steps))
        _cosine_decay = 0.5 * (1.0 + math.cos(math.pi * end_of_warmup_progress))
        warmup_lr_multiplier = min_lr_ratio + (1.0 - min_lr_ratio) * _cosine_decay
        return float(restart_step) / float(max(1, restart_warmup_steps)) * warmup_lr_multiplier

    progress = float(_current_step - first_warmup_steps) / float(
        max(1, num_training_steps - first_warmup_steps)
    )
    cosine_decay = 0.5 * (1.0 + math.cos(math.pi * progress))
    return min_lr_ratio + (1.0 - min_lr_ratio) * cosine_decay


# ---------------------------------------------------------------------------
# optimizer-state pruning / reset (K14)
# ---------------------------------------------------------------------------


@torch.no_grad()
def random_pruning_(tensor, prune_ratio):
    """Zero a random `prune_ratio` fraction of `tensor`, in place."""
    mask = torch.rand_like(tensor) > prune_ratio
    tensor.mul_(mask)


@torch.no_grad()
def magnitude_pruning_(tensor, prune_ratio):
    """Zero the smallest-|t| `prune_ratio` fraction of `tensor`, in place.

    The quantile threshold is computed in fp32 like the reference
    (training_utils.py:160-170). torch.quantile is limited to ~2^24 input
    elements, so large tensors use a sorted-kthvalue threshold instead.
    """
    magnitude = tensor.abs()
    flat = magnitude.flatten().to(torch.float32)
    if flat.numel() < (1 << 24):
        threshold = torch.quantile(flat, prune_ratio)
    else:
        k = max(1, min(flat.numel(), int(round(prune_ratio * (flat.numel() - 1))) + 1))
        threshold = torch.kthvalue(flat, k).values
    mask = magnitude > threshold.to(tensor.dtype)
    tensor.mul_(mask.to(dtype=tensor.dtype))


def optimizer_reset(
    optimizer,
    *,
    reset_params,
    optimizer_state_keys,
    reset_optimizer_on_relora: bool,
    optimizer_random_pruning: float,
    optimizer_magnitude_pruning: float,
):
    """Prune Adam moments of `reset_params` in place after a ReLoRA merge.

    Exactly one mode must be active (reference training_utils.py:267-364).
    """
    n_reset_types = (
        int(bool(reset_optimizer_on_relora))
        + int(bool(optimizer_random_pruning))
        + int(bool(optimizer_magnitude_pruning))
    )
    if n_reset_types != 1:
        logger.warning(
            f"Got {reset_optimizer_on_relora=}, {optimizer_random_pruning=}, {optimizer_magnitude_pruning=}"
        )
        raise ValueError(
            "Exactly one of reset_optimizer_on_relora, optimizer_random_pruning, "
            "optimizer_magnitude_pruning must be True"
        )

    if reset_optimizer_on_relora:
        logger.info("Resetting optimizer states to zeros")
        # 0.999 random-prune instead of hard zero: zeroed-out tensors break
        # ZeRO state_dict consolidation (reference comment training_utils.py:307-346)
        pruning_fn = partial(random_pruning_, prune_ratio=0.999)
    elif optimizer_random_pruning:
        logger.info(f"Performing random pruning of optimizer states. Pruning {optimizer_random_pruning} percent")
        pruning_fn = partial(random_pruning_, prune_ratio=optimizer_random_pruning)
    else:
        logger.info(f"Performing magnitude pruning of optimizer states. Pruning {optimizer_magnitude_pruning} percent")
        pruning_fn = partial(magnitude_pruning_, prune_ratio=optimizer_magnitude_pruning)

    optimizer_state = optimizer.state
    # ZeRO-style wrappers keep the real state on an inner optimizer
    if hasattr(optimizer, "optim"):
        optimizer_state = optimizer.optim.state

    n_zeros = 0
    n_total = 0
    for p in reset_params:
        param_state = optimizer_state.get(p, {})
        if len(param_state) == 0:  # not owned by this rank (ZeRO) or no state yet
            continue
        for key in optimizer_state_keys:
            pruning_fn(param_state[key])
            n_total += param_state[key].numel()
            n_zeros += torch.sum(param_state[key] == 0).item()

    _zeroed = n_zeros / (1e-7 + n_total) * 100
    logger.info(f"Percent of optimizer states zeroed: {_zeroed:.2f}")
    return _zeroed


def print_optimizer_state_size(optimizer):
    first_moment_count = 0
    second_moment_count = 0
    optimizer_state = optimizer.state
    if hasattr(optimizer, "optim"):
        optimizer_state = optimizer.optim.state
    for state in optimizer_state.values():
        if len(state) == 0:
            continue
        first_moment_count += torch.numel(state["exp_avg"])
        second_moment_count += torch.numel(state["exp_avg_sq"])
    global_rank = dist.get_rank() if dist.is_initialized() else 0
    print(f"(Rank {global_rank}) Number of floats in the first moment: {first_moment_count / 1e6:.2f}M")
    print(f"(Rank {global_rank}) Number of floats in the second moment: {second_moment_count / 1e6:.2f}M")


def check_lr_and_alert(optimizer, max_lr):
    global_rank = dist.get_rank() if dist.is_initialized() else 0
    lr = optimizer.param_groups[0]["lr"]
    if lr <= max_lr:
        return
    alert_message = f"Optimizer lr after the reset is large. This can lead to instability. Current lr is {lr}"
    logger.warning(alert_message)
    if global_rank == 0:
        wandb.alert(title="Learning rate issue", text=alert_message,
                    level=getattr(wandb, "AlertLevel", None) and wandb.AlertLevel.WARN)


# ---------------------------------------------------------------------------
# misc / checkpoints
# ---------------------------------------------------------------------------


def max_train_tokens_to_number(max_train_tokens):
    if max_train_tokens.endswith("M"):
        return int(max_train_tokens.rstrip("M")) * 1_000_000
    if max_train_tokens.endswith("B"):
        return int(max_train_tokens.rstrip("B")) * 1_000_000_000
    return int(max_train_tokens)


def get_last_training_state(save_dir):
    """Find the latest `model_{update_step}` checkpoint under save_dir."""
    model_dirs = [d for d in os.listdir(save_dir) if d.startswith("model_")]
    if len(model_dirs) == 0:
        logger.warning(f"Save directory {save_dir} exists, but does not contain any models.")
        logger.warning("Starting training from scratch.")
        return None, None
    model_dirs = sorted(model_dirs, key=lambda x: int(x.split("_")[-1]))
    resume_from = os.path.join(save_dir, model_dirs[-1])
    logger.info(f"Restarting training from {resume_from}")
    with open(os.path.join(resume_from, "training_state.json")) as f:
        training_state = json.load(f)
    return training_state, resume_from


def delete_old_checkpoints(save_dir, keep):
    if keep is None:
        return
    checkpoints = [d for d in os.listdir(save_dir) if d.startswith("model_")]
    if len(checkpoints) <= keep:
        return
    checkpoints = sorted(checkpoints, key=lambda x: int(x.split("_")[-1]))
    for checkpoint in checkpoints[:-keep]:
        checkpoint_path = os.path.join(save_dir, checkpoint)
        logger.info(f"Deleting checkpoint {checkpoint_path}")
        shutil.rmtree(checkpoint_path, ignore_errors=True)
