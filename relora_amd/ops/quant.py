"""Blockwise frozen-weight quantization (K15): NF4 and int8.

MI355X-native replacement for the reference's bitsandbytes usage
(reference relora.py:225-238, 277-299): the frozen `W` of a wrapped Linear
lives quantized at rest (4-bit NF4 with fp32 absmax per 64-element block,
or symmetric linear int8 per 256-element block) and is dequantized into a
transient bf16 buffer for each GEMM; the ReLoRA merge runs
dequant -> W += s·BA -> requant (the reference's exact flow).

GPU path: relora_amd/ops/csrc/quantize.hip.  The pure-python reference
implementations here are the CPU path and the kernels' numerics oracle.
"""

import torch
import torch.nn as nn

from relora_amd.ops import hip

NF4_CODE = torch.tensor([
    -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
    -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
    0.07958029955625534, 0.16093020141124725, 0.24611230194568634,
    0.33791524171829224, 0.44070982933044434, 0.5626170039176941,
    0.7229568362236023, 1.0], dtype=torch.float32)

NF4_BLOCK = 64
I8_BLOCK = 256


def quantize_nf4_ref(x):
    """python oracle: x (any shape, even numel) -> (packed uint8, absmax)."""
    flat = x.detach().float().reshape(-1)
    n = flat.numel()
    nblocks = (n + NF4_BLOCK - 1) // NF4_BLOCK
    padded = torch.zeros(nblocks * NF4_BLOCK, dtype=torch.float32)
    padded[:n] = flat
    blocks = padded.view(nblocks, NF4_BLOCK)
    absmax = blocks.abs().amax(dim=1)
    scaled = blocks / absmax.clamp_min(1e-30).unsqueeze(1)
    scaled[absmax == 0] = 0.0
    idx = (scaled.unsqueeze(-1) - NF4_CODE).abs().argmin(dim=-1).to(torch.uint8)
    idx = idx.view(-1)[:n].view(n // 2, 2)
    packed = (idx[:, 0] << 4) | idx[:, 1]
    return packed.contiguous(), absmax


def dequantize_nf4_ref(packed, absmax, n, dtype=torch.float32):
    hi = (packed >> 4).long()
    lo = (packed & 0xF).long()
    out = torch.empty(n, dtype=torch.float32)
    out[0::2] = NF4_CODE[hi]
    out[1::2] = NF4_CODE[lo]
    blocks = torch.arange(n) // NF4_BLOCK
    out = out * absmax[blocks]
    return out.to(dtype)


def quantize_int8_ref(x):
    flat = x.detach().float().reshape(-1)
    n = flat.numel()
    nblocks = (n + I8_BLOCK - 1) // I8_BLOCK
    padded = torch.zeros(nblocks * I8_BLOCK, dtype=torch.float32)
    padded[:n] = flat
    blocks = padded.view(nblocks, I8_BLOCK)
    absmax = blocks.abs().amax(dim=1)
    scale = torch.where(absmax > 0, 127.0 / absmax, torch.zeros_like(absmax))
    q = torch.round(blocks * scale.unsqueeze(1)).clamp(-127, 127).to(torch.int8)
    return q.view(-1)[:n].contiguous(), absmax


def dequantize_int8_ref(q, absmax, n, dtype=torch.float32):
    blocks = torch.arange(n) // I8_BLOCK
    out = q.float() * (absmax[blocks] / 127.0)
    return out.to(dtype)


class QuantizedWeight(nn.Module):
    """A frozen 2-D weight stored blockwise-quantized.

    Quantization happens on first materialization on a CUDA device (CPU
    keeps the python reference path for tests).  `materialize()` returns a
    dense tensor of the original dtype/shape; `requantize_(w)` re-encodes
    after a ReLoRA merge.
    """

    def __init__(self, weight, mode):
        super().__init__()
        if mode not in ("4bit", "8bit"):
            raise ValueError(f"unknown quantize mode {mode!r}")
        self.mode = mode
        self.out_features, self.in_features = weight.shape
        self.numel = weight.numel()
        self.dtype = weight.dtype
        self._encode(weight.detach())

    def _encode(self, w):
        if w.is_cuda:
            wc = w.contiguous().to(torch.bfloat16)
            if self.mode == "4bit":
                q, absmax = hip.ext().quantize_nf4(wc.view(-1))
            else:
                q, absmax = hip.ext().quantize_int8(wc.view(-1))
        else:
            if self.mode == "4bit":
                q, absmax = quantize_nf4_ref(w)
            else:
                q, absmax = quantize_int8_ref(w)
        self.register_buffer("qdata", q)
        self.register_buffer("absmax", absmax)

    def materialize(self, dtype=None):
        dtype = dtype or self.dtype
        if self.qdata.is_cuda:
            tdtype = torch.bfloat16 if dtype in (torch.bfloat16, torch.float16) else torch.float32
            if self.mode == "4bit":
                flat = hip.ext().dequantize_nf4(self.qdata, self.absmax, self.numel, tdtype)
            else:
                flat = hip.ext().dequantize_int8(self.qdata, self.absmax, self.numel, tdtype)
        else:
            if self.mode == "4bit":
                flat = dequantize_nf4_ref(self.qdata, self.absmax, self.numel, dtype)
            else:
                flat = dequantize_int8_ref(self.qdata, self.absmax, self.numel, dtype)
        return flat.view(self.out_features, self.in_features).to(dtype)

    @torch.no_grad()
    def requantize_(self, w):
        device = self.qdata.device
        self._encode(w.to(device))

    def extra_repr(self):
        return (f"out_features={self.out_features}, in_features={self.in_features}, "
                f"mode={self.mode}")


# absmax must survive model-wide dtype casts (a .to(bf16) on the model would
# otherwise downcast the fp32 scales)
def _quantized_apply(self, fn, recurse=True):
    out = nn.Module._apply(self, fn, recurse)
    if self.absmax.dtype != torch.float32:
        self.absmax = self.absmax.float()
    return out


QuantizedWeight._apply = _quantized_apply
