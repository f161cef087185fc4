"""Pretraining runtime: CLI parsing, distributed setup, the ReLoRA training
loop, eval, and checkpoint/resume.

CLI + behavior parity with the reference `torchrun_main.py` (flag surface
:54-140; main flow :338-1018; train loop :768-947; ReLoRA merge/reset window
:870-916; NaN-consensus batch skipping :810-822; checkpoint layout
`{save_dir}/model_{update_step}/` with pytorch_model.bin + config.json +
relora_config.json + optimizer.pt + training_state.json, :192-225, 830-852).

MI355X-native differences:
* DDP is our `relora_amd.parallel.DistributedModel` — bucketed async
  all-reduce over RCCL/xGMI, overlapped with backward, reduced at
  gradient-accumulation boundaries only;
* `adam_zero` is our `ZeroRedundancyAdamW` (ZeRO-1) with flat-buffer
  parameter broadcasts;
* AdamW and grad clipping run as fused multi-tensor HIP kernels on GPU;
* `--synthetic_data true` (offline extension) trains on deterministic
  random tokens — no network in this environment;
* runs on CPU (gloo) for development/tests when no GPU is present.
"""

import argparse
import json
import os
import random
import time

import numpy as np
import torch
import torch.distributed as dist
import torch.utils.data
import yaml

from relora_amd import args_utils, training_utils
from relora_amd.data.dataloader import SkipDataLoader, SyntheticDataset
from relora_amd.models import build_model_from_config, load_model_config
from relora_amd.models.llama import LlamaForCausalLM
from relora_amd.models.pythia import GPTNeoXForCausalLM
from relora_amd.ops.optim import AdamW, clip_grad_norm_
from relora_amd.ops.tunable import enable_tuned_gemms
from relora_amd.parallel import DistributedModel, ZeroRedundancyAdamW
from relora_amd.relora import ReLoRaLinear, ReLoRaModel
from relora_amd.utils.logging import logger
from relora_amd.utils.wandb_shim import wandb

try:
    from tqdm import tqdm
except ImportError:  # pragma: no cover
    tqdm = None


def default_data_collator(features):
    batch = {}
    for k in features[0].keys():
        vals = [f[k] for f in features]
        if torch.is_tensor(vals[0]):
            batch[k] = torch.stack(vals)
        else:
            batch[k] = torch.tensor(vals)
    return batch


def _int_or_auto(v):
    return v if v == "auto" else int(v)


def parse_args(args=None):
    parser = argparse.ArgumentParser()

    parser.add_argument("--training_config", type=str, default=None,
                        help="Path to a yaml file with training run config. Overrides all other parameters.")

    parser.add_argument("--model_config", type=str, default=None)
    parser.add_argument("--model_name_or_path", type=str, default=None,
                        help="Local model directory (offline), alternative to --model_config")
    parser.add_argument("--model_revision", type=str, default=None)
    parser.add_argument("--warmed_up_model", type=str, default=None,
                        help="Start with warmed-up model weights. Does not restore optimizer and scheduler.")
    parser.add_argument("--resume_from", type=str, default=None,
                        help="Continue training, loading optimizer and scheduler from the checkpoint.")
    parser.add_argument("--load_optimizer_state_on_resume", default=True,
                        type=lambda x: x.lower() == "true")

    parser.add_argument("--dataset_path", type=str, default=None,
                        help="Path to a pre-tokenized huggingface dataset directory")
    parser.add_argument("--megatron_dataset_config", type=str, default=None)
    parser.add_argument("--synthetic_data", default=None, type=lambda x: x.lower() == "true" or None,
                        help="Train on deterministic random tokens (offline benchmarking)")
    parser.add_argument("--max_length", type=int, default=512)

    parser.add_argument("--batch_size", type=_int_or_auto, default=None,
                        help="micro-batch per GPU, or 'auto' to size for the device HBM (288 GB on MI355X)")
    parser.add_argument("--gradient_accumulation", type=int, default=None)
    parser.add_argument("--total_batch_size", type=int, default=None)

    parser.add_argument("--use_peft", default=False, type=lambda x: x.lower() == "true")
    parser.add_argument("--lora_r", type=int, default=128)
    parser.add_argument("--lora_alpha", type=float, default=32)
    parser.add_argument("--relora", type=int, default=None)
    parser.add_argument("--train_scaling", default=False, action="store_true")
    parser.add_argument("--reset_optimizer_on_relora", default=True, type=lambda x: x.lower() == "true")
    parser.add_argument("--optimizer_random_pruning", default=0.0, type=float)
    parser.add_argument("--optimizer_magnitude_pruning", default=0.0, type=float)
    parser.add_argument("--force_keep_original", default=False, type=lambda x: x.lower() == "true")

    parser.add_argument("--optimizer", default="Adam")
    parser.add_argument("--lr", type=float, default=1e-4)
    parser.add_argument("--scheduler", type=str, default="cosine",
                        choices=["linear", "cosine", "cosine_restarts"])
    parser.add_argument("--cycle_length", type=int, default=None)
    parser.add_argument("--restart_warmup_steps", type=int, default=None)
    parser.add_argument("--adjust_step", type=int, default=0)
    parser.add_argument("--min_lr_ratio", type=float, default=0.1)
    parser.add_argument("--adam_beta1", type=float, default=0.9)
    parser.add_argument("--adam_beta2", type=float, default=0.999)
    parser.add_argument("--weight_decay", type=float, default=0.0)
    parser.add_argument("--warmup_steps", type=int, default=1_000)
    parser.add_argument("--clip_grad_norm", type=float, default=1.0)

    parser.add_argument("--eval_every", type=int, default=1_000)
    parser.add_argument("--num_training_steps", type=int, default=10_000)
    parser.add_argument("--max_train_tokens", type=training_utils.max_train_tokens_to_number,
                        default=None)
    parser.add_argument("--save_every", type=int, default=10_000)
    parser.add_argument("--save_dir", type=str, default=None)
    parser.add_argument("--keep_checkpoints", type=int, default=None)
    parser.add_argument("--tags", type=str, default=None)
    parser.add_argument("--dtype", type=str,
                        default="bfloat16" if torch.cuda.is_available() else "float32")
    parser.add_argument("--workers", type=int, default=8)

    parser.add_argument("--quantize", default=None, type=str, choices=[None, "4bit", "8bit"])
    parser.add_argument("--use_double_quant", default=True, type=lambda x: x.lower() == "true")

    parser.add_argument("--distributed_type", type=str, default="ddp", choices=["fsdp", "ddp"])
    parser.add_argument("--profile", default=False, type=lambda x: x.lower() == "true")
    parser.add_argument("--autoresume", default=False, type=lambda x: x.lower() == "true")
    parser.add_argument("--comment", type=str, default=None)
    parser.add_argument("--wandb_watch", default=False, type=lambda x: x.lower() == "true")
    parser.add_argument("--skip_batches", default=None, type=str)

    parser.add_argument("--seed", type=int, default=0)

    n_cli = len(args) if args is not None else None
    args = parser.parse_args(args)
    args = args_utils.check_args_torchrun_main(args, n_cli_args=n_cli)
    return args


@torch.no_grad()
def evaluate_model(model, eval_dataloader, device, target_eval_tokens=10_000_000):
    _time = time.time()
    was_training = model.training
    model.eval()

    ddp_loss_info = torch.zeros(3).to(device)  # [loss, n_batches, n_tokens]
    tokens_in_batch_info = torch.zeros(1).to(device)

    rank = dist.get_rank() if dist.is_initialized() else 0
    n_eval_iters = None
    for i, batch in enumerate(eval_dataloader):
        if i == 0:
            tokens_in_batch_info[0] += batch["input_ids"].numel()
            if dist.is_initialized():
                dist.all_reduce(tokens_in_batch_info, op=dist.ReduceOp.SUM)  # C4
            n_eval_iters = int(target_eval_tokens / tokens_in_batch_info[0])
        if target_eval_tokens != -1 and i > n_eval_iters:
            break
        batch = {k: v.to(device) for k, v in batch.items()}
        loss = model(**batch, labels=batch["input_ids"]).loss
        ddp_loss_info[0] += loss.detach()
        ddp_loss_info[1] += 1
        ddp_loss_info[2] += batch["input_ids"].numel()

    if torch.isnan(ddp_loss_info[0]):
        raise RuntimeError(f"Rank {rank} got nan loss. This is probably a bug.")

    if dist.is_initialized():
        dist.all_reduce(ddp_loss_info, op=dist.ReduceOp.SUM)  # C5
    eval_loss = ddp_loss_info[0] / ddp_loss_info[1]
    evaluated_on_tokens = ddp_loss_info[2].item()
    logger.info(f"Evaluated on {evaluated_on_tokens} tokens, eval loss: {eval_loss:.4f}")
    logger.info(f"Evaluation took {time.time() - _time:.2f} seconds")
    if was_training:
        model.train()
    return eval_loss, evaluated_on_tokens


def save_model(model, *, optimizer, scheduler, training_state_checkpoint, run_config,
               save_dir, dtype):
    """Reference checkpoint layout (save_model_ddp, torchrun_main.py:192-225)."""
    global_rank = dist.get_rank() if dist.is_initialized() else 0
    _time = time.time()

    if global_rank == 0:
        os.makedirs(os.path.dirname(save_dir) or ".", exist_ok=True)
        _model = model.module if hasattr(model, "module") else model
        if isinstance(_model, ReLoRaModel):
            _model.save_pretrained(save_dir)
        else:
            from relora_amd.utils.checkpoint import save_pretrained_compat
            save_pretrained_compat(_model, save_dir)

    if dist.is_initialized():
        dist.barrier()  # C6
    if isinstance(optimizer, ZeroRedundancyAdamW):
        logger.info("Consolidating ZeRO optimizer state dict")
        optimizer.consolidate_state_dict()  # C9

    # per-rank RNG streams so resume is bit-exact even with dropout active
    # (the reference does not restore RNG state across resume; we do better)
    from relora_amd.ops import functional as _ops_functional
    rng_states = {
        "torch": torch.get_rng_state(),
        "numpy": np.random.get_state(),
        "python": random.getstate(),
        "philox_dropout": _ops_functional.get_dropout_rng_state(),
    }
    if torch.cuda.is_available():
        rng_states["cuda"] = torch.cuda.get_rng_state()
    if dist.is_initialized():
        all_rng_states = [None] * dist.get_world_size()
        dist.all_gather_object(all_rng_states, rng_states)
    else:
        all_rng_states = [rng_states]

    if global_rank == 0:
        optimizer_checkpoint = {
            "optimizer": optimizer.state_dict(),
            "scheduler": scheduler.state_dict(),
            "update_step": training_state_checkpoint["update_step"],
            "global_step": training_state_checkpoint["global_step"],
            "config": run_config,
            "dtype": dtype,
            "rng_states_per_rank": all_rng_states,
        }
        torch.save(optimizer_checkpoint, f"{save_dir}/optimizer.pt")
        training_state_checkpoint["wandb_id"] = wandb.run.id if wandb.run else None
        with open(f"{save_dir}/training_state.json", "w") as f:
            json.dump(training_state_checkpoint, f, indent=4)

    logger.info(f"Saving took {time.time() - _time:.2f} seconds")
    if dist.is_initialized():
        dist.barrier()


def maybe_make_profiler(args):
    if not args.profile:
        return None
    global_rank = dist.get_rank() if dist.is_initialized() else 0
    profiler_logging_dir = os.path.join(f"profiler_logs/{args.run_name}")
    prof = torch.profiler.profile(
        schedule=torch.profiler.schedule(wait=1, warmup=1, active=3, repeat=2),
        on_trace_ready=torch.profiler.tensorboard_trace_handler(
            profiler_logging_dir, worker_name=f"rank{global_rank}"
        ),
        record_shapes=True,
        profile_memory=True,
        with_stack=True,
    )
    logger.info(f"Rank {global_rank} profiling results will be saved to {profiler_logging_dir}")
    prof.start()
    return prof


def _save_pretrained_plain(model, save_dir):
    os.makedirs(save_dir, exist_ok=True)
    model.save_pretrained(save_dir, safe_serialization=False)


def main(args):
    torch.manual_seed(args.seed)
    np.random.seed(args.seed)
    random.seed(args.seed)

    global_rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", 1))

    # debug mode (SURVEY.md §5 race-detection plan): serialize kernel
    # launches/copies so HIP faults surface at the offending launch
    if os.environ.get("RELORA_AMD_DEBUG_SERIALIZE") == "1":
        os.environ.setdefault("AMD_SERIALIZE_KERNEL", "3")
        os.environ.setdefault("AMD_SERIALIZE_COPY", "3")
        logger.warning("RELORA_AMD_DEBUG_SERIALIZE: serializing HIP kernels/copies")

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
        enable_tuned_gemms()
        device = f"cuda:{local_rank}"
        backend = "nccl"  # = RCCL on ROCm
    else:
        device = "cpu"
        backend = "gloo"

    logger.info(f"Global rank {global_rank}, local rank {local_rank}, device: {device}")

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group(backend=backend, rank=global_rank, world_size=world_size)  # C1

    if args.batch_size == "auto":
        # size the micro-batch for this device's HBM (288 GB on MI355X)
        from relora_amd.utils.memory import auto_micro_batch

        if args.model_config is None:
            raise ValueError("--batch_size auto requires --model_config")
        _cfg = load_model_config(args.model_config)
        args.batch_size = auto_micro_batch(
            _cfg, args.max_length, args.total_batch_size, world_size,
            lora_r=(args.lora_r or 0) if args.use_peft else 0,
            trainable_ratio=0.08 if args.use_peft else 1.0)
        args.gradient_accumulation = None

    if args.total_batch_size is not None and args.gradient_accumulation is None:
        assert args.total_batch_size % world_size == 0, "total_batch_size must be divisible by world_size"
        args.gradient_accumulation = args.total_batch_size // (args.batch_size * world_size)
        assert args.gradient_accumulation > 0

    assert args.gradient_accumulation * args.batch_size * world_size == args.total_batch_size, \
        "gradient_accumulation * batch_size * world_size must be equal to total_batch_size"

    if args.max_train_tokens is not None:
        args.num_training_steps = args.max_train_tokens // args.total_batch_size

    if global_rank != 0:
        logger.remove()

    # ---- autoresume scan --------------------------------------------------
    wandb_id = None
    if args.save_dir is not None and os.path.exists(args.save_dir):
        if not args.autoresume:
            raise ValueError(
                f"Save directory {args.save_dir} already exists and --autoresume is off. Interrupting..."
            )
        training_state, resume_from = training_utils.get_last_training_state(args.save_dir)
        if args.resume_from is None:
            args.resume_from = resume_from
        if training_state is not None:
            wandb_id = training_state.get("wandb_id")
        logger.info(f"Resuming training from {resume_from} with wandb id {wandb_id}")

    if dist.is_initialized():
        dist.barrier()

    if global_rank == 0:
        wandb.init(project="peft_pretraining", tags=args.tags, id=wandb_id,
                   resume="allow", notes=args.comment,
                   dir=args.save_dir if args.save_dir else None)
        args.run_name = wandb.run.name
        if args.save_dir is None:
            args.save_dir = f"checkpoints/{wandb.run.name}"
        os.makedirs(args.save_dir, exist_ok=True)
        with open(os.path.join(args.save_dir, "training_config.yaml"), "w") as f:
            yaml.dump({k: v for k, v in vars(args).items() if k != "skip_batches"}, f)

    if dist.is_initialized():
        dist.barrier()
        # C7 + save_dir: both are derived on rank 0 (save_dir from the run
        # name when not given) — replicate so every rank formats the same
        # checkpoint paths
        meta = [wandb.run.name, args.save_dir] if global_rank == 0 else ["", None]
        dist.broadcast_object_list(meta, src=0)
        args.run_name, args.save_dir = meta
    else:
        args.run_name = getattr(args, "run_name", "local")
    if args.save_dir is None:
        args.save_dir = f"checkpoints/{args.run_name}"

    logger.info("*" * 40)
    for k, v in vars(args).items():
        logger.info(f"{k:30} {v}")
    logger.info("*" * 40)

    # ---- data -------------------------------------------------------------
    test_loader = None
    train_loader = eval_loader = None
    train_dataset = eval_dataset = None
    dataset_preprocessing_args = {"tokenizer": "synthetic"}
    tokenizer = None

    if args.dataset_path is not None:
        import datasets
        import datasets.distributed

        logger.info("Loading Huggingface dataset from directory")
        dataset_dict = datasets.load_from_disk(args.dataset_path)
        dataset_dict.set_format(type="torch", columns=["input_ids"])
        train_dataset = dataset_dict["train"]
        if args.seed != 0:
            train_dataset = train_dataset.shuffle(seed=args.seed)
        eval_dataset = dataset_dict["validation"]

        minimum_n_tokens = args.total_batch_size * args.num_training_steps
        dataset_n_tokens = len(train_dataset) * args.max_length
        if dataset_n_tokens < minimum_n_tokens:
            raise ValueError(
                f"Dataset only has {dataset_n_tokens} tokens, but we need at least {minimum_n_tokens}"
            )
        with open(os.path.join(args.dataset_path, "args.json")) as f:
            dataset_preprocessing_args = json.load(f)
        assert dataset_preprocessing_args["sequence_length"] == args.max_length
        try:
            from transformers import AutoTokenizer
            tokenizer = AutoTokenizer.from_pretrained(
                dataset_preprocessing_args["tokenizer"], model_max_length=args.max_length
            )
        except Exception as e:  # offline: tokenizer only needed for vocab check
            logger.warning(f"Could not load tokenizer ({e}); skipping vocab check")
            tokenizer = None

    elif args.megatron_dataset_config is not None:
        from relora_amd.data.megatron import load_megatron_dataset
        start_iteration = 0
        if args.model_revision is not None and args.model_revision.startswith("step"):
            start_iteration = int(args.model_revision[4:])
        train_loader, eval_loader, test_loader, tokenizer = load_megatron_dataset(
            args, world_size=world_size, start_iteration=start_iteration
        )
        dataset_preprocessing_args = {"tokenizer": getattr(tokenizer, "name_or_path", "megatron")}

    # ---- model ------------------------------------------------------------
    if args.model_config is not None:
        model_config = load_model_config(args.model_config)
        if tokenizer is not None:
            t_vocab = tokenizer.get_vocab_size() if hasattr(tokenizer, "get_vocab_size") \
                else tokenizer.vocab_size
            if model_config.vocab_size != t_vocab:
                logger.warning(
                    f"Model config vocab size ({model_config.vocab_size}) does not match "
                    f"tokenizer vocab size ({t_vocab})"
                )
                if not (model_config.vocab_size == 32000 and t_vocab == 32100):
                    raise ValueError("Model config vocab size does not match tokenizer vocab size")
        model = build_model_from_config(model_config)
    else:
        logger.info(f"Loading local model from {args.model_name_or_path}")
        model_config = load_model_config(args.model_name_or_path)
        model = build_model_from_config(model_config)
        state_path = os.path.join(args.model_name_or_path, "pytorch_model.bin")
        if os.path.exists(state_path):
            model.load_state_dict(
                torch.load(state_path, map_location="cpu", weights_only=True), strict=True
            )

    if args.synthetic_data:
        n_train = max(args.total_batch_size * (args.num_training_steps + 1), args.total_batch_size)
        train_dataset = SyntheticDataset(model_config.vocab_size, args.max_length,
                                         n_train, seed=args.seed + 1)
        eval_dataset = SyntheticDataset(model_config.vocab_size, args.max_length,
                                        max(64, args.batch_size * 4), seed=args.seed + 2)

    global_step = 0
    update_step = 0
    tokens_seen = 0
    tokens_seen_before = 0
    n_lora_restarts = 0
    n_optimizer_resets = 0

    if args.warmed_up_model is not None:
        logger.info(f"Loading a warmed-up model from {args.warmed_up_model}")
        checkpoint_path = os.path.join(args.warmed_up_model, "pytorch_model.bin")
        model.load_state_dict(
            torch.load(checkpoint_path, map_location="cpu", weights_only=True), strict=True
        )
        ts_path = os.path.join(args.warmed_up_model, "training_state.json")
        if os.path.exists(ts_path):
            with open(ts_path) as f:
                _old_state = json.load(f)
            global_step = _old_state["global_step"]
            update_step = _old_state["update_step"]
            tokens_seen = _old_state["tokens_seen"]
            tokens_seen_before = _old_state["tokens_seen_before"]
            logger.info(f"Warm start: update_step={update_step}, tokens_seen={tokens_seen}")

    params_before = sum(p.numel() for p in model.parameters())

    if args.use_peft:
        need_linear_weight = (
            args.relora is not None or args.force_keep_original
            or args.warmed_up_model is not None
        )
        logger.info(f"Wrapping model with LoRA ({need_linear_weight=})")
        model = ReLoRaModel(
            model,
            r=args.lora_r,
            lora_alpha=args.lora_alpha,
            lora_dropout=0.1,
            target_modules=["attn", "attention", "mlp"],
            trainable_scaling=args.train_scaling,
            keep_original_weights=True,
            lora_only=not need_linear_weight,
            quantize=args.quantize,
            use_double_quant=args.use_double_quant,
        )

    if args.resume_from:
        logger.info(f"Loading model from {args.resume_from}")
        checkpoint_path = os.path.join(args.resume_from, "pytorch_model.bin")
        sd = torch.load(checkpoint_path, map_location="cpu", weights_only=True)
        if isinstance(model, ReLoRaModel):
            model.wrapped_model.load_state_dict(sd, strict=True)
        else:
            model.load_state_dict(sd, strict=True)
        with open(os.path.join(args.resume_from, "training_state.json")) as f:
            _old_state = json.load(f)
        global_step = _old_state["global_step"]
        _update_step = _old_state["update_step"]
        tokens_seen = _old_state["tokens_seen"]
        tokens_seen_before = _old_state["tokens_seen_before"]
        n_lora_restarts = _old_state.get("n_lora_restarts", 0)
        n_optimizer_resets = _old_state.get("n_optimizer_resets", 0)
        logger.info(f"Resume: global_step={global_step}, update_step={_update_step}")
        if args.megatron_dataset_config is not None and hasattr(train_loader, "batch_sampler"):
            train_loader.batch_sampler.start_iter = global_step

    params_after = sum(p.numel() for p in model.parameters())

    logger.info(f"Total params  before LoRA: {params_before / 1e6:.2f}M")
    logger.info(f"Total params  after  LoRA: {params_after / 1e6:.2f}M")
    logger.info(
        f"Trainable params: {sum(p.numel() for p in model.parameters() if p.requires_grad) / 1e6:.2f}M"
    )

    if args.dtype in ["bf16", "bfloat16"]:
        model = model.to(device=device, dtype=torch.bfloat16)
    else:
        model = model.to(device=device)

    n_total_params = sum(p.numel() for p in model.parameters())
    n_trainable_params = sum(p.numel() for p in model.parameters() if p.requires_grad)
    p_trainable_params = n_trainable_params / n_total_params

    # ---- distributed wrapping (our DDP-equivalent) ------------------------
    model = DistributedModel(model)

    trainable_params = [p for p in model.parameters() if p.requires_grad]
    lora_params = [p for n, p in model.named_parameters() if p.requires_grad and "lora_" in n]
    trainable_params_names = [n for n, p in model.named_parameters() if p.requires_grad]

    if args.use_peft and len(lora_params) == 0:
        raise ValueError("No LoRA parameters found")

    run_config = dict(vars(args))
    run_config.pop("skip_batches", None)
    run_config.update({
        "tokenizer": dataset_preprocessing_args.get("tokenizer"),
        "max_lr": run_config.pop("lr"),
        "total_params_M": n_total_params / 1e6,
        "trainable_params_M": n_trainable_params / 1e6,
        "equivalent_params_M": params_before / 1e6,
        "percent_trainable_params": p_trainable_params,
        "name_trainable_params": trainable_params_names,
        "model": model_config.to_dict(),
        "world_size": world_size,
        "device": str(device),
        "dataset_preprocessing_args": dataset_preprocessing_args,
    })
    if global_rank == 0:
        wandb.config.update(run_config, allow_val_change=True)

    optimizer_state_keys = ["exp_avg", "exp_avg_sq"]
    optimizer_kwargs = {
        "lr": args.lr,
        "weight_decay": args.weight_decay,
        "betas": (args.adam_beta1, args.adam_beta2),
    }
    if args.optimizer.lower() == "adam":
        optimizer = AdamW(trainable_params, **optimizer_kwargs)
    elif args.optimizer.lower() == "adam_zero":
        optimizer = ZeroRedundancyAdamW(trainable_params, **optimizer_kwargs)
    else:
        raise ValueError(f"Optimizer {args.optimizer} not supported")

    scheduler_start_step = update_step
    scheduler = training_utils.get_scheculer(
        optimizer=optimizer if not isinstance(optimizer, ZeroRedundancyAdamW) else optimizer.optim,
        scheduler_type=args.scheduler,
        num_training_steps=args.num_training_steps - scheduler_start_step,
        warmup_steps=args.warmup_steps,
        min_lr_ratio=args.min_lr_ratio,
        cycle_length=args.cycle_length,
        restart_warmup_steps=args.restart_warmup_steps,
        adjust_step=args.adjust_step,
    )

    if args.resume_from:
        for _ in range(update_step):
            scheduler.step()
        if args.load_optimizer_state_on_resume:
            optimizer_checkpoint = torch.load(
                os.path.join(args.resume_from, "optimizer.pt"), map_location="cpu",
                weights_only=False,
            )
            optimizer.load_state_dict(optimizer_checkpoint["optimizer"])
            scheduler.load_state_dict(optimizer_checkpoint["scheduler"])
            update_step = optimizer_checkpoint["update_step"]
            global_step = optimizer_checkpoint["global_step"]
            _rng = optimizer_checkpoint.get("rng_states_per_rank")
            if _rng and len(_rng) != world_size:
                logger.warning(
                    f"Checkpoint holds RNG streams for {len(_rng)} ranks but "
                    f"world_size is {world_size}; bit-exact resume is only "
                    "guaranteed at the original world size (extra ranks keep "
                    "fresh seeds / streams are reused by position).")
            if _rng and global_rank < len(_rng):
                _st = _rng[global_rank]
                torch.set_rng_state(_st["torch"])
                np.random.set_state(_st["numpy"])
                random.setstate(_st["python"])
                if "cuda" in _st and torch.cuda.is_available():
                    torch.cuda.set_rng_state(_st["cuda"])
                if "philox_dropout" in _st:
                    from relora_amd.ops import functional as _ops_functional
                    _ops_functional.set_dropout_rng_state(_st["philox_dropout"])
            logger.info(f"Optimizer and scheduler restored from {args.resume_from}")
        _tc_path = os.path.join(args.resume_from, "training_config.yaml")
        if os.path.exists(_tc_path):
            with open(_tc_path) as f:
                _old_training_config = yaml.safe_load(f)
            if args.batch_size != _old_training_config.get("batch_size"):
                raise RuntimeError("Cannot resume from a checkpoint with a different batch size.")

    # ---- dataloaders -------------------------------------------------------
    # Every DataLoader gets a dedicated seeded generator: DataLoader.__iter__
    # draws one int64 from the GLOBAL RNG for its _base_seed, so without this
    # the global stream would be offset by where/when iterators are created —
    # which breaks bit-exact resume (a fresh run creates the train iterator at
    # step 0, a resumed run at the resume step; everything downstream of the
    # merge re-init would then diverge).
    def _loader_gen(salt):
        return torch.Generator().manual_seed(args.seed * 100003 + salt)

    if args.dataset_path is not None:
        import datasets.distributed
        train_dataset = datasets.distributed.split_dataset_by_node(
            train_dataset, rank=global_rank, world_size=world_size)
        eval_dataset = datasets.distributed.split_dataset_by_node(
            eval_dataset, rank=global_rank, world_size=world_size)
        _skip_batches = update_step * args.gradient_accumulation
        train_loader = SkipDataLoader(
            train_dataset, batch_size=args.batch_size, collate_fn=default_data_collator,
            skip_batches=_skip_batches, num_workers=args.workers,
            generator=_loader_gen(1),
        )
        eval_loader = torch.utils.data.DataLoader(
            eval_dataset, batch_size=args.batch_size, collate_fn=default_data_collator,
            num_workers=args.workers, generator=_loader_gen(2),
        )
    elif args.synthetic_data:
        # deterministic contiguous shard per rank
        sampler = torch.utils.data.distributed.DistributedSampler(
            train_dataset, num_replicas=world_size, rank=global_rank, shuffle=False,
        ) if world_size > 1 else None
        _skip = update_step * args.gradient_accumulation
        train_loader = SkipDataLoader(
            train_dataset, batch_size=args.batch_size, collate_fn=default_data_collator,
            skip_batches=_skip, num_workers=min(args.workers, 2), sampler=sampler,
            generator=_loader_gen(1),
        )
        eval_loader = torch.utils.data.DataLoader(
            eval_dataset, batch_size=args.batch_size, collate_fn=default_data_collator,
            num_workers=0, generator=_loader_gen(2),
        )
    else:
        assert train_loader is not None and eval_loader is not None

    update_time = time.time()
    local_step = 0
    loss_info = torch.tensor([0.0, 0.0, 0.0], device=device)
    n_skipped_batches = 0

    prof = maybe_make_profiler(args)

    logger.info(
        f"Starting training at update step {update_step} "
        f"({args.num_training_steps - update_step} update steps to go)"
    )
    pbar = None
    if global_rank == 0 and tqdm is not None and os.environ.get("RELORA_AMD_NO_TQDM") != "1":
        pbar = tqdm(total=args.num_training_steps - update_step, desc="Update steps", ncols=80)

    for batch in train_loader:
        global_step += 1
        local_step += 1

        if update_step in args.skip_batches:
            if global_step % args.gradient_accumulation == 0:
                update_step += 1
            continue

        if update_step >= args.num_training_steps:
            logger.info(f"Reached max number of update steps ({args.num_training_steps}). Stopping training.")
            break

        batch = {k: v.to(device) for k, v in batch.items()}
        tokens_seen += batch["input_ids"].numel() * world_size

        is_boundary = global_step % args.gradient_accumulation == 0
        model.set_gradient_sync(is_boundary)  # reduce only at accumulation boundary

        loss = model(**batch, labels=batch["input_ids"]).loss

        loss_info[0] += loss.detach()
        loss_info[1] += 1
        loss_info[2] += torch.isnan(loss).float()

        scaled_loss = loss / args.gradient_accumulation
        scaled_loss.backward()

        if not is_boundary:
            continue

        # ----- update step -----
        model.finish_gradient_sync()
        if pbar is not None:
            pbar.update(1)

        if args.clip_grad_norm > 0:
            grad_norm = clip_grad_norm_(trainable_params, args.clip_grad_norm,
                                        error_if_nonfinite=True)
            if global_rank == 0:
                wandb.log({"grad_norm": float(grad_norm)}, step=global_step)

        if dist.is_initialized():
            dist.all_reduce(loss_info, op=dist.ReduceOp.SUM)  # C3: NaN consensus
        _loss = loss_info[0] / loss_info[1]

        if loss_info[2] == 0:
            optimizer.step()
            scheduler.step()
        else:
            logger.error(f"Nan detected in loss_info, {_loss=}, skipping update")
            n_skipped_batches += 1
            if n_skipped_batches > 0.05 * args.num_training_steps:
                logger.error("More than 5% of batches skipped due to NaNs, stopping training.")
                break

        model.zero_grad_buffers()
        update_step += 1
        update_time = time.time() - update_time
        loss_info = torch.zeros_like(loss_info)

        if local_step > args.gradient_accumulation and update_step % args.save_every == 0:
            current_model_directory = f"{args.save_dir}/model_{update_step}"
            training_state_checkpoint = {
                "global_step": global_step,
                "update_step": update_step,
                "tokens_seen": tokens_seen,
                "tokens_seen_before": tokens_seen_before,
                "n_lora_restarts": n_lora_restarts,
                "n_optimizer_resets": n_optimizer_resets,
                "update_time": update_time,
            }
            save_model(
                model, optimizer=optimizer, scheduler=scheduler,
                training_state_checkpoint=training_state_checkpoint,
                run_config=run_config, save_dir=current_model_directory, dtype=args.dtype,
            )
            if args.keep_checkpoints is not None and global_rank == 0:
                training_utils.delete_old_checkpoints(args.save_dir, keep=args.keep_checkpoints)

        if update_step % args.eval_every == 0:
            total_loss, evaluated_on_tokens = evaluate_model(model, eval_loader, device)
            if global_rank == 0:
                wandb.log({"final_eval_loss": float(total_loss),
                           "final_eval_tokens": evaluated_on_tokens}, step=global_step)
            logger.info(f"Eval loss at step {update_step}: {total_loss}")

        # ----- ReLoRA merge + optimizer reset window (reference :870-916) -----
        can_reset_relora = args.relora is not None and (
            args.resume_from is not None
            or local_step // args.gradient_accumulation >= args.relora
        )
        if can_reset_relora and (update_step - scheduler_start_step) % args.relora == 1:
            logger.info(f"Performing lora reset at update step {update_step}. "
                        f"Current lr is {optimizer.param_groups[0]['lr']}")
            n_lora_restarts += 1
            model.module.merge_and_reinit()

        can_reset_optimizer = args.relora is not None and (
            args.resume_from is not None
            or local_step // args.gradient_accumulation >= args.cycle_length
        )
        if can_reset_optimizer and (update_step - scheduler_start_step) % args.cycle_length == 1:
            logger.info(f"Performing optimizer reset at update step {update_step}. "
                        f"Current lr is {optimizer.param_groups[0]['lr']}")
            n_optimizer_resets += 1
            training_utils.optimizer_reset(
                optimizer,
                reset_params=lora_params,
                optimizer_state_keys=optimizer_state_keys,
                reset_optimizer_on_relora=args.reset_optimizer_on_relora,
                optimizer_random_pruning=args.optimizer_random_pruning,
                optimizer_magnitude_pruning=args.optimizer_magnitude_pruning,
            )
        if can_reset_optimizer and (update_step - scheduler_start_step) % args.cycle_length == 2:
            logger.info(f"First step after optimizer reset lr is {optimizer.param_groups[0]['lr']}")

        lr = optimizer.param_groups[0]["lr"]
        tokens_in_update = tokens_seen - tokens_seen_before
        tokens_seen_before = tokens_seen
        batches_in_update = args.gradient_accumulation * world_size

        if global_rank == 0:
            wandb.log({
                "loss": float(_loss),
                "lr": lr,
                "update_step": update_step,
                "tokens_seen": tokens_seen,
                "throughput_tokens": tokens_in_update / update_time,
                "throughput_examples": args.total_batch_size / update_time,
                "throughput_batches": batches_in_update / update_time,
                "n_lora_restarts": n_lora_restarts,
                "n_optimizer_resets": n_optimizer_resets,
            }, step=global_step)
            if args.train_scaling:
                all_scaling_factors = [
                    m.scaling.data.item() for m in model.modules()
                    if isinstance(m, ReLoRaLinear)
                ]
                wandb.log({"lora_scaling": all_scaling_factors}, step=global_step)
        update_time = time.time()
        if prof is not None:
            prof.step()
    else:
        logger.warning("Reached the end of the dataset. Training stopped")

    if prof is not None:
        prof.stop()
    logger.info("Training finished")
    if pbar is not None:
        pbar.close()

    current_model_directory = f"{args.save_dir}/model_{update_step}"
    # The reference guards the final save with a per-rank os.path.exists
    # (torchrun_main.py:956) — a cross-rank race: rank 0 creates the
    # directory inside save_model while a slower rank then sees it existing,
    # skips the save (and its barrier), and the job deadlocks between
    # save_model's barrier and the final eval's all_reduce.  Decide once on
    # rank 0 and broadcast so every rank takes the same branch.
    need_final_save = [not os.path.exists(current_model_directory)]
    if dist.is_initialized() and world_size > 1:
        dist.broadcast_object_list(need_final_save, src=0)
    if need_final_save[0]:
        training_state_checkpoint = {
            "global_step": global_step,
            "update_step": update_step,
            "tokens_seen": tokens_seen,
            "tokens_seen_before": tokens_seen_before,
            "n_lora_restarts": n_lora_restarts,
            "n_optimizer_resets": n_optimizer_resets,
            "update_time": update_time,
        }
        save_model(
            model, optimizer=optimizer, scheduler=scheduler,
            training_state_checkpoint=training_state_checkpoint,
            run_config=run_config, save_dir=current_model_directory, dtype=args.dtype,
        )

    logger.info("Running final evaluation")
    model.eval()
    if use_gpu:
        torch.cuda.empty_cache()
    total_loss, evaluated_on_tokens = evaluate_model(
        model, eval_loader, device, target_eval_tokens=100_000_000
    )
    if global_rank == 0:
        wandb.log({"final_eval_loss": float(total_loss),
                   "final_eval_tokens": evaluated_on_tokens}, step=global_step)
        logger.info(f"Final eval loss: {total_loss}")

    if test_loader is not None:
        total_loss, evaluated_on_tokens = evaluate_model(
            model, test_loader, device, target_eval_tokens=-1
        )
        if global_rank == 0:
            wandb.log({"final_test_loss": float(total_loss),
                       "final_test_tokens": evaluated_on_tokens}, step=global_step)

    if global_rank == 0:
        wandb.finish()
    logger.info("Script finished successfully")
    print(f"Rank {global_rank} finished successfully")
