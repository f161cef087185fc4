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
    p.add_argument("--suite", action="store_true",
                   help="also bench the secondary configs (llama_250m r=128 "
                        "bs8, llama_7b r=256 bs4) before the flagship; one "
                        "JSON line each, flagship printed LAST")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", 1))

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        # clamp so a world-2-on-one-GPU proof run (RCCL permitting) works
        dev = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev)
        device = f"cuda:{dev}"
        backend = "nccl"  # RCCL on ROCm
    else:
        device = "cpu"
        backend = "gloo"

    # init the process group whenever launched under torchrun (RANK set),
    # including world_size 1: that makes a single-GPU `torchrun
    # --nproc-per-node 1` run exercise RCCL init + the bucketed all-reduce
    # path for real (with RELORA_AMD_FORCE_SYNC=1).
    if (world_size > 1 or "RANK" in os.environ) and not dist.is_initialized():
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
    # fresh synthetic batch per step, pre-generated on device OUTSIDE the
    # timed region, so throughput cannot benefit from single-batch caching
    # effects and loss stays a sanity signal rather than memorization
    g = torch.Generator(device="cpu").manual_seed(42 + rank)
    batches = [
        torch.randint(0, cfg.vocab_size, (B, S), generator=g).to(device)
        for _ in range(args.warmup + args.steps)
    ]
    loss_info = torch.zeros(3, device=device)

    def step(batch):
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

    for i in range(args.warmup):
        step(batches[i])

    # per-step boundaries via CUDA events (no per-step host sync inside the
    # timed region); the headline time stays the single host-clock bracket
    ev = [torch.cuda.Event(enable_timing=True)
          for _ in range(args.steps + 1)] if use_gpu else None

    if dist.is_initialized():
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        if ev:
            ev[i].record()
        loss = step(batches[args.warmup + i])
    if ev:
        ev[args.steps].record()
    if use_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0

    step_ms = ([ev[i].elapsed_time(ev[i + 1]) for i in range(args.steps)]
               if ev else [elapsed / args.steps * 1000] * args.steps)

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
                "step_ms_min": round(min(step_ms), 2),
                "step_ms_max": round(max(step_ms), 2),
                "step_ms_stdev": round(
                    (sum((t - sum(step_ms) / len(step_ms)) ** 2
                         for t in step_ms) / len(step_ms)) ** 0.5, 2),
            },
        }
        print(json.dumps(result))

    if dist.is_initialized():
        dist.destroy_process_group()


def _run_suite():
    """Secondary configs as subprocesses (fresh HIP context each), flagship
    last so last-line JSON parsing still lands on the headline number."""
    import subprocess
    import sys
    base = [a for a in sys.argv[1:] if a != "--suite"]
    for extra in (
        ["--model", "configs/llama_250m.json", "--batch_size", "8"],
        ["--model", "configs/llama_7b.json", "--batch_size", "4",
         "--lora_r", "256"],
    ):
        subprocess.run([sys.executable, __file__] + base + extra, check=False)
    main()


if __name__ == "__main__":
    import sys
    if "--suite" in sys.argv:
        _run_suite()
    else:
        main()
