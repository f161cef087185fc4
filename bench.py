"""Flagship benchmark: Llama-1B ReLoRA r=128 pretraining step throughput.

Measures the BASELINE.json headline metric — tokens/sec for the whole node —
on synthetic data with random-init weights (no network in this environment),
bf16, reference hyperparameters (r=128, alpha=32, dropout 0.1, clip 1.0,
AdamW), seq_len 2048, micro-batch 8 per GPU, one full update step per
measured step (fwd + bwd + overlapped RCCL all-reduce + clip + fused AdamW).

Launch (the driver does this):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-node aggregate.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="configs/llama_1b.json")
    p.add_argument("--model-name", type=str, default=None,
                   help="label for the config block (default: file stem)")
    p.add_argument("--batch_size", type=int, default=8, help="micro-batch per GPU")
    p.add_argument("--seq_len", type=int, default=2048)
    p.add_argument("--lora_r", type=int, default=128)
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--full_rank", action="store_true",
                   help="bench full-rank training instead of ReLoRA")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", 1))

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
        backend = "nccl"  # RCCL on ROCm
    else:
        device = "cpu"
        backend = "gloo"

    if world_size > 1 and not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)

    from relora_amd.models import build_model_from_config, load_model_config
    from relora_amd.ops.tunable import enable_tuned_gemms
    from relora_amd.ops.optim import AdamW, clip_grad_norm_
    from relora_amd.parallel import DistributedModel
    from relora_amd.relora import ReLoRaModel

    torch.manual_seed(1234)
    if use_gpu:
        enable_tuned_gemms()
    cfg = load_model_config(args.model)
    model = build_model_from_config(cfg)
    if not args.full_rank:
        model = ReLoRaModel(
            model, r=args.lora_r, lora_alpha=32, lora_dropout=0.1,
            target_modules=["attn", "attention", "mlp"], keep_original_weights=True,
        )
    dtype = torch.bfloat16 if args.dtype in ("bf16", "bfloat16") else torch.float32
    model = model.to(device=device, dtype=dtype)
    model = DistributedModel(model)
    model.train()

    trainable = [p for p in model.parameters() if p.requires_grad]
    n_total = sum(p.numel() for p in model.parameters())
    n_trainable = sum(p.numel() for p in trainable)
    optimizer = AdamW(trainable, lr=4e-4, betas=(0.9, 0.95), weight_decay=0.01)

    B, S = args.batch_size, args.seq_len
    g = torch.Generator(device="cpu").manual_seed(42 + rank)
    batch = torch.randint(0, cfg.vocab_size, (B, S), generator=g).to(device)
    loss_info = torch.zeros(3, device=device)

    def step():
        model.set_gradient_sync(True)
        loss = model(input_ids=batch, labels=batch).loss
        loss_info[0] = loss.detach()
        loss_info[1] = 1
        loss_info[2] = torch.isnan(loss.detach()).float()
        loss.backward()
        model.finish_gradient_sync()
        clip_grad_norm_(trainable, 1.0, error_if_nonfinite=True)
        if dist.is_initialized():
            dist.all_reduce(loss_info, op=dist.ReduceOp.SUM)
        optimizer.step()
        model.zero_grad_buffers()
        return loss

    for _ in range(args.warmup):
        step()

    if dist.is_initialized():
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    if use_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_per_step = B * S * world_size
    value = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000

    if rank == 0:
        model_name = args.model_name or os.path.splitext(os.path.basename(args.model))[0]
        result = {
            "metric": "throughput_tokens (tokens/sec, node)",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # the reference publishes no numbers
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": B * world_size,
                "seq_len": S,
                "parallelism": f"dp{world_size}",
                "relora_r": None if args.full_rank else args.lora_r,
                "total_params_M": round(n_total / 1e6, 1),
                "trainable_params_M": round(n_trainable / 1e6, 1),
                "percent_trainable": round(100 * n_trainable / n_total, 1),
                "final_loss": round(float(loss.detach()), 4),
            },
        }
        print(json.dumps(result))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
