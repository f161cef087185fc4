"""Peak-HBM A/B: quantized (--quantize 4bit) vs dense bf16 llama ReLoRA
training step.  The VERDICT K15 criterion: quantized must train in
materially less HBM than bf16 (round 1 materialized dense W per forward
and saved it for backward, erasing the saving).

  python tools/bench_quant_mem.py [--model configs/llama_1b.json] [--bs 4]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch


def run(model_cfg, bs, seq, quantize):
    from relora_amd.models import build_model_from_config, load_model_config
    from relora_amd.ops.optim import AdamW
    from relora_amd.relora import ReLoRaModel

    torch.manual_seed(0)
    cfg = load_model_config(model_cfg)
    model = build_model_from_config(cfg)
    model = ReLoRaModel(model, r=128, lora_alpha=32, lora_dropout=0.1,
                        target_modules=["attn", "attention", "mlp"],
                        keep_original_weights=True, quantize=quantize)
    model = model.to("cuda", dtype=torch.bfloat16)
    trainable = [p for p in model.parameters() if p.requires_grad]
    opt = AdamW(trainable, lr=1e-4)
    batch = torch.randint(0, cfg.vocab_size, (bs, seq), device="cuda")
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    t0 = time.perf_counter()
    for _ in range(3):
        loss = model(input_ids=batch, labels=batch).loss
        loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=False)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    peak = torch.cuda.max_memory_allocated() / 2**30
    del model, opt
    torch.cuda.empty_cache()
    return peak, dt, float(loss)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="configs/llama_1b.json")
    p.add_argument("--bs", type=int, default=4)
    p.add_argument("--seq", type=int, default=2048)
    args = p.parse_args()
    for q in [None, "4bit"]:
        peak, dt, loss = run(args.model, args.bs, args.seq, q)
        print(f"quantize={q}: peak HBM {peak:.2f} GiB, {dt*1e3:.0f} ms/step, "
              f"loss {loss:.3f}")


if __name__ == "__main__":
    main()
