"""ReLoRA adapter: LoRA A/B on every targeted Linear with periodic
merge-and-reinit into the frozen weight.

API parity with the reference (`peft_pretraining/relora.py`):
`ReLoRaConfig` (:18-28), `ReLoRaModel` (:49-177, module surgery :94-134,
zero-init equivalence :120-124, save/from_pretrained :149-177),
`ReLoRaLinear` (:181-323, merge_and_reinit :269-307, forward :309-323) and
`merge_and_reinit_functional` (:31-46).

MI355X notes:
* the forward routes through `relora_amd.ops.lora_linear`, whose GPU path
  fuses the rank-r update into the main MFMA GEMM (ops/csrc/lora_gemm.hip);
* merge `W += B·A·scale` runs in the model dtype, exactly like the reference
  (bf16 accumulation per cycle — documented deviation point in SURVEY.md §7
  hard-part 4);
* the 4-bit/8-bit quantized frozen-W path replaces bitsandbytes with our
  blockwise NF4/int8 HIP kernels (relora_amd/ops/quant.py,
  ops/csrc/quantize.hip): W lives quantized at rest, dequantizes per GEMM,
  and the merge runs dequant -> += BA·s -> requant.
"""

import json
import math
import os
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from relora_amd import ops
from relora_amd.utils.logging import logger


@dataclass
class ReLoRaConfig:
    r: int
    lora_alpha: int
    lora_dropout: float
    target_modules: List[str]
    keep_original_weights: bool
    lora_only: bool = False
    trainable_scaling: bool = False
    quantize: Optional[str] = None
    use_double_quant: bool = False


def merge_and_reinit_functional(module):
    """Out-of-class merge (kept for FSDP-style use, reference relora.py:31-46)."""
    if not isinstance(module, ReLoRaLinear):
        return
    delta = module.lora_B.weight @ module.lora_A.weight
    delta = delta * module._post_lora_scale()
    if module.quantize is None:
        module.weight.data += delta
    else:
        w = module.weight.materialize(module.lora_A.weight.dtype)
        module.weight.requantize_(w + delta)
    nn.init.kaiming_uniform_(module.lora_A.weight, a=math.sqrt(5))
    nn.init.zeros_(module.lora_B.weight)
    if module.trainable_scaling:
        nn.init.zeros_(module.scaling)


class ReLoRaModel(torch.nn.Module):
    """Wraps every nn.Linear whose qualified name contains a target key in a
    ReLoRaLinear (frozen W + trainable low-rank A/B)."""

    def __init__(
        self,
        model,
        *,
        target_modules,
        r=128,
        lora_alpha=32,
        lora_dropout=0.1,
        keep_original_weights=True,
        lora_only=False,
        trainable_scaling=False,
        quantize=None,
        use_double_quant=False,
    ):
        if r <= 0:
            raise ValueError("LoRA rank r must be >= 1; for a rank-0 (plain) layer just skip the ReLoRA wrap.")
        super().__init__()
        self.wrapped_model: nn.Module = model
        self.r = r
        self.lora_alpha = lora_alpha
        self.lora_dropout = lora_dropout
        self.target_modules = target_modules
        self.keep_original_weights = keep_original_weights
        self.lora_only = lora_only
        self.trainable_scaling = trainable_scaling

        self._config = ReLoRaConfig(
            r=r,
            lora_alpha=lora_alpha,
            lora_dropout=lora_dropout,
            target_modules=target_modules,
            keep_original_weights=keep_original_weights,
            lora_only=lora_only,
            trainable_scaling=trainable_scaling,
            quantize=quantize,
            use_double_quant=use_double_quant,
        )

        # expose the wrapped model's forward (reference relora.py:89) and,
        # beyond the reference, its generation surface so a wrapped model
        # can decode directly
        self.forward = self.wrapped_model.forward
        if hasattr(self.wrapped_model, "generate"):
            self.generate = self.wrapped_model.generate

        target_modules_list = [target_modules] if isinstance(target_modules, str) else target_modules

        to_replace = []
        for module_name, module in self.wrapped_model.named_modules():
            if not isinstance(module, nn.Linear):
                continue
            if not any(key in module_name for key in target_modules_list):
                continue
            to_replace.append((module_name, module))

        for module_name, module in to_replace:
            weight_data = module.weight.data if keep_original_weights else None
            bias_data = None
            if module.bias is not None:
                bias_data = module.bias.data if keep_original_weights else None

            new_module = ReLoRaLinear(
                module.in_features,
                module.out_features,
                r=self.r,
                bias=module.bias is not None,
                lora_alpha=self.lora_alpha,
                lora_dropout=self.lora_dropout,
                lora_only=self.lora_only,
                trainable_scaling=self.trainable_scaling,
                quantize=quantize,
                weight_data=weight_data,
                bias_data=bias_data,
            )
            if self.keep_original_weights:
                # the wrapped network is exactly the original at init:
                # lora_A is zeroed so B·A ≡ 0 (reference relora.py:120-124)
                nn.init.zeros_(new_module.lora_A.weight)
                assert new_module.lora_A.bias is None
                assert new_module.lora_B.bias is None
            if self.lora_only:
                assert not self.keep_original_weights
                module.weight = None

            parent_name = ".".join(module_name.split(".")[:-1])
            parent = self.wrapped_model.get_submodule(parent_name)
            setattr(parent, module_name.split(".")[-1], new_module)

        if torch.cuda.is_available():
            torch.cuda.empty_cache()

    @torch.no_grad()
    def merge_and_reinit(self):
        for module in self.modules():
            if isinstance(module, ReLoRaLinear):
                module.merge_and_reinit()

    def save_pretrained(self, path):
        # reference layout: pytorch_model.bin (+ config.json) + relora_config.json
        from relora_amd.utils.checkpoint import save_pretrained_compat
        save_pretrained_compat(self.wrapped_model, path)
        with open(os.path.join(path, "relora_config.json"), "w") as f:
            json.dump(self._config.__dict__, f, indent=4)

    @classmethod
    def from_pretrained(cls, path):
        from relora_amd.models import build_model_from_config
        from relora_amd.models.config import load_model_config

        with open(os.path.join(path, "relora_config.json")) as f:
            relora_config = json.load(f)

        config = load_model_config(path)
        base_model = build_model_from_config(config)

        if "keep_original" in relora_config:  # legacy key (reference relora.py:162-166)
            relora_config["lora_only"] = not relora_config.pop("keep_original")
            relora_config["keep_original_weights"] = not relora_config["lora_only"]
        relora_config.setdefault("trainable_scaling", False)

        model = cls(base_model, **relora_config)

        from relora_amd.utils.checkpoint import load_state_dict_compat
        model.wrapped_model.load_state_dict(load_state_dict_compat(path), strict=True)
        return model


class ReLoRaLinear(nn.Module):
    """x ↦ x Wᵀ (+b) + s · dropout(x) Aᵀ Bᵀ with frozen W and s = α/r
    (or tanh(scaling) when trainable)."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        r: int,
        *,
        lora_alpha: int = 1,
        lora_dropout: float = 0.1,
        lora_only: bool = False,
        weight_data=None,
        bias_data=None,
        trainable_scaling: bool = False,
        bias=True,
        device=None,
        dtype=None,
        quantize=None,
    ):
        super().__init__()
        if r <= 0:
            raise ValueError("LoRA rank r must be >= 1; for a rank-0 (plain) layer just skip the ReLoRA wrap.")
        if quantize not in (None, "4bit", "8bit"):
            raise ValueError(f"quantize must be None, '4bit' or '8bit', got {quantize!r}")

        if lora_only:
            self.weight = None
            self.bias = None
        else:
            if bias_data is None:
                bias_data = (
                    torch.zeros(out_features, device=device, dtype=dtype, requires_grad=True)
                    if bias else None
                )
            self.bias = nn.Parameter(bias_data) if bias else None
            if weight_data is None:
                weight_data = torch.zeros(
                    out_features, in_features, device=device, dtype=dtype, requires_grad=False
                )
            if quantize is None:
                self.weight = nn.Parameter(weight_data, requires_grad=False)
            else:
                # frozen W lives blockwise-quantized (NF4 / int8) and is
                # dequantized per GEMM (reference's bitsandbytes flow,
                # relora.py:225-238 — ours is relora_amd/ops/quant.py)
                from relora_amd.ops.quant import QuantizedWeight

                self.weight = QuantizedWeight(weight_data, quantize)

        self.in_features = in_features
        self.out_features = out_features
        self.r = r
        self.lora_alpha = lora_alpha
        self.lora_dropout_p = lora_dropout
        self.lora_dropout = nn.Dropout(p=lora_dropout)  # kept for module-dict parity
        self.lora_only = lora_only
        self.trainable_scaling = trainable_scaling
        self.quantize = quantize

        self.lora_A = nn.Linear(in_features, r, bias=False)
        nn.init.kaiming_uniform_(self.lora_A.weight, a=math.sqrt(5))
        self.lora_B = nn.Linear(r, out_features, bias=False)
        nn.init.zeros_(self.lora_B.weight)
        if trainable_scaling:
            self.scaling = nn.Parameter(torch.tensor([1.0]), requires_grad=True)
        else:
            self.scaling = self.lora_alpha / self.r
        if not self.lora_only and isinstance(self.weight, nn.Parameter):
            self.weight.requires_grad = False

    def _post_lora_scale(self):
        if self.trainable_scaling:
            return self.scaling.tanh()
        return self.scaling

    def _dense_weight(self):
        """The frozen W as a dense tensor (dequantized when quantize is set)."""
        if self.quantize is None:
            return self.weight
        return self.weight.materialize(self.lora_A.weight.dtype)

    @torch.no_grad()
    def merge_and_reinit(self):
        if self.lora_only:
            logger.warning("Skipping merge and reinit, because only lora parameters are used")
            return
        if self.quantize is None:
            merged = False
            if (self.weight.is_cuda and self.weight.dtype == torch.bfloat16
                    and self.r % 32 == 0 and self.r <= 256
                    and not torch.is_tensor(self._post_lora_scale())):
                from relora_amd.ops import hip as _hip
                if _hip.use_hip(self.weight, "merge"):
                    # K13 on the MFMA accumulate kernel: W += (B*s) @ A
                    # in-place (fp32 accumulation, single bf16 rounding —
                    # the torch form rounds the product before the add)
                    bs = (self.lora_B.weight * self._post_lora_scale()).contiguous()
                    _hip.ext().lora_add_nn_(
                        self.weight.data, bs, self.lora_A.weight.contiguous(),
                        bs.new_empty(0, dtype=torch.uint8), 1.0)
                    merged = True
            if not merged:
                self.weight.data += self.lora_B.weight @ self.lora_A.weight * self._post_lora_scale()
        else:
            # dequant -> merge -> requant (reference relora.py:277-299)
            w = self.weight.materialize(self.lora_A.weight.dtype)
            w += self.lora_B.weight @ self.lora_A.weight * self._post_lora_scale()
            self.weight.requantize_(w)
        nn.init.kaiming_uniform_(self.lora_A.weight, a=math.sqrt(5))
        nn.init.zeros_(self.lora_B.weight)
        if self.trainable_scaling:
            nn.init.zeros_(self.scaling)

    def forward(self, x: torch.Tensor):
        # quantized: pass the packed weight through — the GPU path runs the
        # dequant-fused GEMM and never holds a dense [out,in] W (K15); the
        # CPU path materializes transiently inside lora_linear
        quantized = self.weight if self.quantize is not None else None
        return ops.lora_linear(
            x,
            self.weight if self.quantize is None else None,
            self.bias,
            self.lora_A.weight,
            self.lora_B.weight,
            self._post_lora_scale(),
            dropout_p=self.lora_dropout_p,
            training=self.training,
            lora_only=self.lora_only,
            quantized_weight=quantized,
        )

    def extra_repr(self):
        return (f"in_features={self.in_features}, out_features={self.out_features}, "
                f"r={self.r}, lora_alpha={self.lora_alpha}")
