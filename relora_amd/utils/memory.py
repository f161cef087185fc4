"""Micro-batch auto-sizing for 288 GB HBM3E per MI355X GPU.

`--batch_size auto` picks the largest micro-batch whose estimated training
footprint fits the device (north-star requirement: size for 288 GB, fewer
and larger collectives).  The model is analytic — parameters, gradients,
Adam states, logits buffers and the dominant per-layer activations saved
for backward — with a safety margin; it intentionally under-fills rather
than OOMs mid-run.
"""

import torch

from relora_amd.utils.logging import logger

_SAFETY = 0.85  # fraction of free HBM the estimate may claim


def _bytes_per_dtype(dtype):
    return torch.tensor([], dtype=dtype).element_size()


def estimate_step_bytes(cfg, micro_batch, seq_len, dtype=torch.bfloat16,
                        trainable_ratio=1.0, lora_r=0):
    """Rough peak bytes for one training step at the given micro-batch."""
    e = _bytes_per_dtype(dtype)
    H = cfg.hidden_size
    L = cfg.num_hidden_layers
    I = cfg.intermediate_size
    V = cfg.vocab_size
    nh = cfg.num_attention_heads
    T = micro_batch * seq_len

    n_params = V * H * 2 + L * (4 * H * H + 3 * H * I + 2 * H) + H
    lora_params = L * lora_r * (4 * (H + H) + 2 * (H + I) + (I + H)) if lora_r else 0
    weights = (n_params + lora_params) * e
    # grads + two Adam states for the trainable set
    n_train = int(n_params * trainable_ratio) + lora_params
    opt = n_train * e * 3

    # activations saved for backward, per layer:
    #  norms in/out (2H), qkv+rope (3H), attention out (H), o-in (H),
    #  mlp gate/up/act (2I + I), down-in (I) ≈ 7H + 4I per token
    act_layer = T * (7 * H + 4 * I) * e
    # attention saves q,k,v,o + lse
    act_attn = T * (4 * H) * e + T * nh * 4
    acts = L * (act_layer + act_attn)
    # logits path: bf16 logits + grad (chunked CE still peaks at one chunk)
    ce = 2 * min(T, 16384) * V * e
    embeds = 3 * T * H * e
    return weights + opt + acts + ce + embeds


def auto_micro_batch(cfg, seq_len, total_batch_size, world_size,
                    dtype=torch.bfloat16, lora_r=0, trainable_ratio=0.05,
                    hbm_bytes=None):
    """Largest power-of-two micro-batch that fits and divides the global batch."""
    if hbm_bytes is None:
        if torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info()
            hbm_bytes = free
        else:
            hbm_bytes = 16 << 30  # CPU testing default
    budget = hbm_bytes * _SAFETY
    per_rank_cap = max(1, total_batch_size // world_size)
    bs = 1
    # only grow while the result still divides the per-rank batch, so the
    # trainer's `grad_accum * bs * world == total_batch_size` algebra holds
    # (e.g. per_rank_cap=24 stops at 8, not 16)
    while bs * 2 <= per_rank_cap and per_rank_cap % (bs * 2) == 0:
        need = estimate_step_bytes(cfg, bs * 2, seq_len, dtype,
                                   trainable_ratio, lora_r)
        if need > budget:
            break
        bs *= 2
    need = estimate_step_bytes(cfg, bs, seq_len, dtype, trainable_ratio, lora_r)
    logger.info(
        f"auto micro-batch: {bs} (est. {need / 1e9:.1f} GB of "
        f"{hbm_bytes / 1e9:.1f} GB free, seq {seq_len})")
    return bs
