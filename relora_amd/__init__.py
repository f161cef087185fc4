"""relora_amd — an MI355X-native (CDNA4/gfx950) ReLoRA pretraining engine.

Re-implements the capabilities of the reference ReLoRA codebase
(`Guitaricet/relora`) with an AMD-first compute path:

* hand-written HIP kernels (MFMA / LDS-tiled for gfx950) for the hot ops —
  fused LoRA GEMM, flash-style causal attention, RMSNorm, RoPE, fused
  cross-entropy, multi-tensor AdamW / grad-clip, merge_and_reinit — see
  ``relora_amd.ops``;
* RCCL-over-xGMI data parallelism with bucketed, backward-overlapped gradient
  all-reduce and optional ZeRO-1 optimizer sharding — see
  ``relora_amd.parallel``;
* the reference's public API surface: ``ReLoRaModel`` / ``ReLoRaLinear``
  (relora.py), ``get_scheculer`` / ``optimizer_reset`` (training_utils.py),
  the Llama / Pythia model family (models/), the HF and Megatron data paths
  (data/), and the ``torchrun_main.py`` CLI + checkpoint layout.
"""

__version__ = "0.1.0"

from relora_amd.relora import (  # noqa: F401
    ReLoRaConfig,
    ReLoRaLinear,
    ReLoRaModel,
    merge_and_reinit_functional,
)
