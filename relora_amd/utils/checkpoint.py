"""Checkpoint I/O in the reference's on-disk layout.

The reference saves HF-style directories containing `pytorch_model.bin` +
`config.json` (reference relora.py:149-152, torchrun_main.py:200-222).
transformers 5.x `save_pretrained` writes safetensors only, so we write the
layout ourselves (and read either format back)."""

import os

import torch


def save_pretrained_compat(model, save_dir):
    """Write `config.json` + `pytorch_model.bin` for an HF-style model."""
    os.makedirs(save_dir, exist_ok=True)
    if hasattr(model, "config"):
        model.config.save_pretrained(save_dir)
    torch.save(model.state_dict(), os.path.join(save_dir, "pytorch_model.bin"))


# Buffer keys the reference persists that our models recompute (fp32 RoPE
# caches are non-persistent here; the attention bias mask is implicit in the
# flash kernel).  Dropped on load so reference-written checkpoints load
# strict (reference modeling_llama.py rotary_emb.inv_freq,
# modeling_pythia.py attention.bias/masked_bias + cos/sin caches).
_LEGACY_BUFFER_KEYS = ("rotary_emb.inv_freq", "attention.bias",
                       "attention.masked_bias", "cos_cached", "sin_cached")


def _drop_legacy_buffers(sd):
    return {k: v for k, v in sd.items()
            if not any(k.endswith(s) for s in _LEGACY_BUFFER_KEYS)}


def load_state_dict_compat(path):
    """Load a state dict from `pytorch_model.bin` or `model.safetensors`,
    dropping the reference's persistent rotary/mask buffers."""
    bin_path = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(bin_path):
        return _drop_legacy_buffers(
            torch.load(bin_path, map_location="cpu", weights_only=True))
    st_path = os.path.join(path, "model.safetensors")
    if os.path.exists(st_path):
        from safetensors.torch import load_file
        return _drop_legacy_buffers(load_file(st_path))
    raise FileNotFoundError(f"No model weights found under {path}")
