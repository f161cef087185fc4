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


def load_state_dict_compat(path):
    """Load a state dict from `pytorch_model.bin` or `model.safetensors`."""
    bin_path = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(bin_path):
        return torch.load(bin_path, map_location="cpu", weights_only=True)
    st_path = os.path.join(path, "model.safetensors")
    if os.path.exists(st_path):
        from safetensors.torch import load_file
        return load_file(st_path)
    raise FileNotFoundError(f"No model weights found under {path}")
