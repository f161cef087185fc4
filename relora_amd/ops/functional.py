"""Functional compute ops with HIP/CDNA4 dispatch.

Each op has two implementations:

* a hand-written gfx950 HIP kernel (``relora_amd/ops/csrc``, loaded via
  :mod:`relora_amd.ops.hip`) — the production path on MI355X;
* a pure-PyTorch implementation — the CPU path and the numerics oracle the
  kernels are tested against (tests/test_ops_*.py).

Kernel inventory parity: K1/K2 `lora_linear` (fused LoRA
GEMM), K3 `flash_attention`, K4 `rope`, K5 `rmsnorm`, K6 `layernorm`,
K7 `swiglu`, K10 `fused_cross_entropy` (chunked; never materializes *fp32*
logits — row stats and the in-place grad are fp32-accurate HIP kernels over
bf16 logit chunks. The default single-chunk config does hold one full
[M,V] bf16 logits buffer (~1.6 GB at the flagship shape — cheap against
288 GB HBM); set RELORA_AMD_CE_CHUNK lower to bound that. The reference
flags this memory hot spot at modeling_llama.py:696-697),
K11/K12 live in :mod:`relora_amd.ops.optim`,
K13 `merge_and_reinit` in :mod:`relora_amd.relora`, K14 pruning in
:mod:`relora_amd.training_utils`.
"""

import os

import torch
import torch.nn.functional as F

from relora_amd.ops import hip

# ---------------------------------------------------------------------------
# RMSNorm (K5) — reference numerics: fp32 variance, bf16 product
# (reference modeling_llama.py:74-91)
# ---------------------------------------------------------------------------


def rmsnorm_torch(x, weight, eps):
    variance = x.to(torch.float32).pow(2).mean(-1, keepdim=True)
    h = x * torch.rsqrt(variance + eps)
    if weight.dtype in (torch.float16, torch.bfloat16):
        h = h.to(weight.dtype)
    return weight * h


class _HipRMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        x2d = x.contiguous().view(-1, x.shape[-1])
        y, invrms = hip.ext().rmsnorm_fwd(x2d, weight, eps)
        ctx.save_for_backward(x2d, weight, invrms)
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, invrms = ctx.saved_tensors
        dy2d = dy.contiguous().view(-1, dy.shape[-1])
        dx, dw = hip.ext().rmsnorm_bwd(x2d, weight, invrms, dy2d)
        return dx.view(ctx.shape), dw.to(weight.dtype), None


def rmsnorm(x, weight, eps=1e-6):
    if hip.use_hip(x, "rmsnorm"):
        return _HipRMSNorm.apply(x, weight, eps)
    return rmsnorm_torch(x, weight, eps)


class _HipAddRMSNorm(torch.autograd.Function):
    """Fused `sum = x + residual; normed = rmsnorm(sum)` (K16).  Backward
    returns the SAME gradient tensor for both inputs (d sum/dx = d sum/dres
    = 1) with the residual fork's `+dsum` folded into the dx epilogue — the
    unfused form costs one [M,H] add kernel in each direction.

    Note: because the two returned gradients alias one tensor, calling this
    on two LEAF tensors makes their `.grad`s aliases as well (fine for
    values; an in-place edit of one would show in the other).  The model
    only ever feeds activations, whose gradients are consumed, not stored.
    """

    @staticmethod
    def forward(ctx, x, residual, weight, eps):
        shape = x.shape
        x2d = x.contiguous().view(-1, shape[-1])
        r2d = residual.contiguous().view(-1, shape[-1])
        y, s, invrms = hip.ext().rmsnorm_fwd_add(x2d, r2d, weight, eps)
        ctx.save_for_backward(s, weight, invrms)
        ctx.shape = shape
        return y.view(shape), s.view(shape)

    @staticmethod
    def backward(ctx, dy, dsum):
        s, weight, invrms = ctx.saved_tensors
        H = s.shape[-1]
        dy2d = dy.contiguous().view(-1, H)
        if dsum is None:
            dx, dw = hip.ext().rmsnorm_bwd(s, weight, invrms, dy2d)
        else:
            dx, dw = hip.ext().rmsnorm_bwd_add(s, weight, invrms, dy2d,
                                               dsum.contiguous().view(-1, H))
        dx = dx.view(ctx.shape)
        return dx, dx, dw.to(weight.dtype), None


def add_rmsnorm(x, residual, weight, eps=1e-6):
    """(normed, sum) where sum = x + residual, normed = rmsnorm(sum).
    Same bf16 rounding as the unfused composition."""
    if hip.use_hip(x, "rmsnorm") and x.dtype == torch.bfloat16 \
            and residual.dtype == torch.bfloat16:
        return _HipAddRMSNorm.apply(x, residual, weight, eps)
    s = x + residual
    return rmsnorm(s, weight, eps), s


# ---------------------------------------------------------------------------
# LayerNorm (K6) — pythia path
# ---------------------------------------------------------------------------


class _HipLayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x2d = x.contiguous().view(-1, x.shape[-1])
        y, mean, invstd = hip.ext().layernorm_fwd(x2d, weight, bias, eps)
        ctx.save_for_backward(x2d, weight, mean, invstd)
        ctx.shape = x.shape
        return y.view(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, mean, invstd = ctx.saved_tensors
        dy2d = dy.contiguous().view(-1, dy.shape[-1])
        dx, dw, db = hip.ext().layernorm_bwd(x2d, weight, mean, invstd, dy2d)
        return dx.view(ctx.shape), dw.to(weight.dtype), db.to(weight.dtype), None


def layernorm(x, weight, bias, eps=1e-5):
    if hip.use_hip(x, "layernorm"):
        return _HipLayerNorm.apply(x, weight, bias, eps)
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


# ---------------------------------------------------------------------------
# RoPE (K4) — rotate_half convention (reference modeling_llama.py:126-141),
# partial-rotary support for pythia (modeling_pythia.py:184-197)
# ---------------------------------------------------------------------------


def rotate_half(x):
    x1 = x[..., : x.shape[-1] // 2]
    x2 = x[..., x.shape[-1] // 2 :]
    return torch.cat((-x2, x1), dim=-1)


def build_rope_cache(rot_dim, seq_len, base=10000.0, device=None):
    """fp32 cos/sin tables of shape [seq_len, rot_dim] (duplicated halves),
    identical to the reference's `emb = cat((freqs, freqs))` layout."""
    inv_freq = 1.0 / (base ** (torch.arange(0, rot_dim, 2, dtype=torch.float32, device=device) / rot_dim))
    t = torch.arange(seq_len, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return emb.cos(), emb.sin()


def rope_torch(q, k, cos, sin, position_ids=None):
    """q,k: [B, nh, S, hd]; cos/sin: [S_cache, rot_dim] fp32.

    Rotates the first rot_dim features, passes the rest through.
    """
    rot = cos.shape[-1]
    S = q.shape[-2]
    if position_ids is None:
        c = cos[:S].to(q.dtype)
        s = sin[:S].to(q.dtype)
    else:
        c = cos[position_ids].to(q.dtype).unsqueeze(1)
        s = sin[position_ids].to(q.dtype).unsqueeze(1)
    q_rot, q_pass = q[..., :rot], q[..., rot:]
    k_rot, k_pass = k[..., :rot], k[..., rot:]
    q_out = (q_rot * c) + (rotate_half(q_rot) * s)
    k_out = (k_rot * c) + (rotate_half(k_rot) * s)
    if q_pass.shape[-1]:
        q_out = torch.cat((q_out, q_pass), dim=-1)
        k_out = torch.cat((k_out, k_pass), dim=-1)
    return q_out, k_out


class _HipRoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        # the kernel reads BHSD-contiguous AND transposed-BSHD views directly
        # (same stride support as attention — no copies on the q/k path)
        if not (_attn_layout_ok(q) and q.stride() == k.stride()):
            q, k = q.contiguous(), k.contiguous()
        # Kernel contract: fp32 contiguous tables (a model .to(bf16) casts
        # registered buffers; converting [S,hd] back is noise next to the GEMMs).
        cos = cos.float().contiguous()
        sin = sin.float().contiguous()
        qo, ko = hip.ext().rope_fwd(q, k, cos, sin, False)
        ctx.save_for_backward(cos, sin)
        return qo, ko

    @staticmethod
    def backward(ctx, dqo, dko):
        cos, sin = ctx.saved_tensors
        if not (_attn_layout_ok(dqo) and dqo.stride() == dko.stride()):
            dqo, dko = dqo.contiguous(), dko.contiguous()
        dq, dk = hip.ext().rope_fwd(dqo, dko, cos, sin, True)
        return dq, dk, None, None


def rope(q, k, cos, sin, position_ids=None):
    """Apply rotary embedding to q and k. position_ids only supported on the
    torch path (training uses the contiguous [0..S) default)."""
    if position_ids is None and hip.use_hip(q, "rope"):
        return _HipRoPE.apply(q, k, cos, sin)
    return rope_torch(q, k, cos, sin, position_ids)


# ---------------------------------------------------------------------------
# SwiGLU (K7)
# ---------------------------------------------------------------------------


def swiglu_torch(gate, up):
    return F.silu(gate) * up


class _HipSwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        gate = gate.contiguous()
        up = up.contiguous()
        y = hip.ext().swiglu_fwd(gate, up)
        ctx.save_for_backward(gate, up)
        return y

    @staticmethod
    def backward(ctx, dy):
        gate, up = ctx.saved_tensors
        dgate, dup = hip.ext().swiglu_bwd(gate, up, dy.contiguous())
        return dgate, dup


def swiglu(gate, up):
    if hip.use_hip(gate, "swiglu"):
        return _HipSwiGLU.apply(gate, up)
    return swiglu_torch(gate, up)


# ---------------------------------------------------------------------------
# GELU (K8) — pythia MLP, exact erf form (reference modeling_pythia.py:395-406)
# ---------------------------------------------------------------------------


class _HipGELU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.save_for_backward(x)
        return hip.ext().gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return hip.ext().gelu_bwd(x, dy.contiguous())


def gelu(x):
    if hip.use_hip(x, "gelu"):
        return _HipGELU.apply(x)
    return F.gelu(x)


# ---------------------------------------------------------------------------
# Flash attention (K3) — causal, no padding mask (parity with the reference:
# SDPA is_causal=True and padding mask ignored, modeling_llama.py:221-224)
# ---------------------------------------------------------------------------


def sdpa_torch(q, k, v, causal=True, dropout_p=0.0):
    return F.scaled_dot_product_attention(q, k, v, dropout_p=dropout_p, is_causal=causal)


def _attn_layout_ok(t):
    """BHSD-contiguous, or a transposed view of a BSHD buffer (what the
    model's `.view(B,S,nh,hd).transpose(1,2)` produces) — the kernels read
    both directly, so no permute+contiguous copies happen around attention."""
    if t.stride(-1) != 1:
        return False
    B, nh, S, hd = t.shape
    s = tuple(t.stride())
    return (t.is_contiguous()
            or s == (S * nh * hd, hd, nh * hd, 1))


class _HipFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        if not (_attn_layout_ok(q) and q.stride() == k.stride() == v.stride()):
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = hip.ext().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        # the host copies `do` into q's layout only if it differs
        dq, dk, dv = hip.ext().attn_bwd(q, k, v, o, lse, do, ctx.scale)
        return dq, dk, dv, None


def flash_attention(q, k, v, causal=True, dropout_p=0.0, scale=None):
    """q,k,v: [B, nh, S, hd] -> [B, nh, S, hd]."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if causal and dropout_p == 0.0 and hip.use_hip(q, "attention"):
        return _HipFlashAttention.apply(q, k, v, scale)
    return F.scaled_dot_product_attention(
        q, k, v, dropout_p=dropout_p, is_causal=causal, scale=scale
    )


# ---------------------------------------------------------------------------
# Fused chunked cross-entropy over the LM head (K10).
# loss = CE(shift(hidden @ Wᵀ), shift(labels)) without materializing fp32
# logits: per-M-chunk bf16 GEMM + one-pass fp32 row stats (HIP) and an
# in-place softmax-minus-onehot gradient kernel in backward.  Peak logits
# memory is one [chunk, V] bf16 buffer (the full [M, V] only in the default
# single-chunk config, sized for 288 GB HBM — see _CE_CHUNK below).
# ---------------------------------------------------------------------------

# 288 GB HBM comfortably holds one [M,V] bf16 logits buffer for the flagship
# shapes (16384 x 50304 = 1.6 GB), so default to a single chunk: fewer, larger
# GEMMs and no fp32 dw accumulation pass. Shrink via env on smaller cards.
_CE_CHUNK = int(os.environ.get("RELORA_AMD_CE_CHUNK", "16384"))
# RELORA_AMD_CE_SAVE_LOGITS trades HBM for time: keep the bf16 logits from
# forward instead of recomputing the [M,V] GEMM in backward (single-chunk
# mode only; ~1 GB at the flagship shape).  Default ON for 288 GB parts —
# measured +0.4-0.5% e2e on two boxes; =0 to disable on small-HBM cards.
_CE_SAVE_LOGITS = os.environ.get("RELORA_AMD_CE_SAVE_LOGITS", "1") == "1"


def _row_stats_torch(logits, labels, ignore_index):
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    safe = labels.clamp_min(0)
    tgt = lf.gather(1, safe.unsqueeze(1)).squeeze(1)
    return lse, tgt


class _FusedCrossEntropy(torch.autograd.Function):
    """hidden [M,H], weight [V,H], labels [M] -> mean CE over labels != ignore."""

    @staticmethod
    def forward(ctx, hidden, weight, labels, ignore_index):
        M = hidden.shape[0]
        device = hidden.device
        lse_all = torch.empty(M, dtype=torch.float32, device=device)
        loss_sum = torch.zeros((), dtype=torch.float32, device=device)
        valid = labels != ignore_index
        n_valid = int(valid.sum().item())
        use_hip = hip.use_hip(hidden, "ce")
        keep_logits = _CE_SAVE_LOGITS and M <= _CE_CHUNK
        saved_logits = None
        for s in range(0, M, _CE_CHUNK):
            e = min(s + _CE_CHUNK, M)
            logits = hidden[s:e] @ weight.t()
            if keep_logits:
                saved_logits = logits
            lab = labels[s:e]
            if use_hip:
                lse, tgt = hip.ext().ce_row_stats(logits, lab, ignore_index)
            else:
                lse, tgt = _row_stats_torch(logits, lab, ignore_index)
            lse_all[s:e] = lse
            vmask = lab != ignore_index
            loss_sum += torch.where(vmask, lse - tgt, torch.zeros_like(lse)).sum()
        ctx.save_for_backward(hidden, weight, labels, lse_all,
                              saved_logits if saved_logits is not None
                              else hidden.new_empty(0))
        ctx.ignore_index = ignore_index
        ctx.n_valid = max(n_valid, 1)
        return loss_sum / max(n_valid, 1)

    @staticmethod
    def backward(ctx, grad_out):
        hidden, weight, labels, lse_all, saved_logits = ctx.saved_tensors
        ignore_index = ctx.ignore_index
        M, H = hidden.shape
        use_hip = hip.use_hip(hidden, "ce")
        dh = torch.empty_like(hidden)
        single_chunk = M <= _CE_CHUNK
        dw_acc = None if single_chunk else torch.zeros(
            weight.shape, dtype=torch.float32, device=weight.device)
        gscale = (grad_out.float() / ctx.n_valid).item() if grad_out.dim() == 0 else None
        for s in range(0, M, _CE_CHUNK):
            e = min(s + _CE_CHUNK, M)
            if saved_logits.numel() and s == 0 and e == M:
                logits = saved_logits
            else:
                logits = hidden[s:e] @ weight.t()
            lab = labels[s:e]
            lse = lse_all[s:e]
            if use_hip:
                # in-place: logits buffer becomes dlogits (same dtype)
                hip.ext().ce_grad_(logits, lab, lse, gscale, ignore_index)
                dlogits = logits
            else:
                p = torch.exp(logits.float() - lse.unsqueeze(1))
                vmask = (lab != ignore_index)
                p[~vmask] = 0.0
                safe = lab.clamp_min(0)
                p[vmask, safe[vmask]] -= 1.0
                dlogits = (p * gscale).to(logits.dtype)
            dh[s:e] = dlogits @ weight
            if single_chunk:
                dw = (dlogits.t() @ hidden[s:e]).to(weight.dtype)
            else:
                dw_acc += (dlogits.t() @ hidden[s:e]).float()
        if not single_chunk:
            dw = dw_acc.to(weight.dtype)
        return dh, dw, None, None


def fused_cross_entropy(hidden, weight, labels, ignore_index=-100):
    """Mean cross-entropy of `hidden @ weight.T` against `labels`.

    hidden: [M, H] (already shifted/flattened); weight: [V, H]; labels: [M].
    """
    return _FusedCrossEntropy.apply(hidden, weight, labels, ignore_index)


# ---------------------------------------------------------------------------
# LoRA linear (K1+K2): y = x Wᵀ (+b) + s · dropout(x) Aᵀ Bᵀ
# The production GPU path fuses the rank-r update into the main MFMA GEMM
# (lora_gemm.hip); RELORA_AMD_LORA_PATH=torch forces the hipBLASLt
# composition (also the CPU path).
# ---------------------------------------------------------------------------


_lora_seed_counter = [0]


def get_dropout_rng_state():
    """Philox dropout stream position (for bit-exact checkpoint resume)."""
    return _lora_seed_counter[0]


def set_dropout_rng_state(counter):
    _lora_seed_counter[0] = int(counter)


def _next_dropout_seed():
    _lora_seed_counter[0] += 1
    # mix with the torch seed so runs differ when the user reseeds, while
    # successive calls in one run are distinct and deterministic
    return (torch.initial_seed() & 0x7FFFFFFFFFFF) ^ (_lora_seed_counter[0] * 0x9E3779B97F4A7C15)


# The single-kernel fused GEMM+LoRA (ops/csrc/fused_gemm.hip) replaces
# hipBLASLt+lora_add where the measured A/B favors it: aligned shapes with
# K <= 1024 (profiles/fused_gemm_ab.log — fused wins +8..19% on the
# llama_250m shapes; at K >= 2048 the library's deeper-pipelined GEMM wins
# by 6-36%, so those stay composed).  RELORA_AMD_FUSED_K1=0 disables,
# =all forces it on every aligned shape (for A/B runs).
_FUSED_K1 = os.environ.get("RELORA_AMD_FUSED_K1", "1")


def _use_fused_k1(x2d, weight, lora_A):
    if _FUSED_K1 == "0":
        return False
    M, K = x2d.shape
    N = weight.shape[0]
    r = lora_A.shape[0]
    aligned = (M % 256 == 0 and N % 256 == 0 and K % 64 == 0
               and r % 64 == 0 and r <= 256)
    return aligned and (K <= 1024 or _FUSED_K1 == "all")


class _FusedLoRALinear(torch.autograd.Function):
    """GPU path: the rank-r update accumulates into the main GEMM's output
    via the MFMA lora_add kernels; dropout runs fused with a persisted
    packed philox mask that backward replays exactly.  Saves the three
    [M,N]-sized epilogue round trips per Linear of the composed path."""

    @staticmethod
    def forward(ctx, x, weight, bias, lora_A, lora_B, scale, dropout_p, training):
        in_shape = x.shape
        K = in_shape[-1]
        N = weight.shape[0]
        x2d = x.contiguous().view(-1, K)
        use_dropout = dropout_p > 0 and training
        if use_dropout:
            seed = _next_dropout_seed() & 0x7FFFFFFFFFFFFFFF
            xd, mask = hip.ext().dropout_mask_fwd(x2d, dropout_p, seed)
        else:
            xd, mask = x2d, None
        t_u = xd @ lora_A.t()                       # [M, r]
        del xd  # dA re-applies the mask inline in skinny_grad; no need to
                # persist the dropped-out copy (64 MB per flagship linear)
        bs = lora_B * scale                         # [N, r]
        if _use_fused_k1(x2d, weight, lora_A):
            # single-kernel K1+K2: main MFMA GEMM with the rank-r update as
            # an epilogue on the resident accumulators (no [M,N] RMW pass)
            y = hip.ext().fused_lora_gemm(
                x2d, weight, t_u, lora_B,
                bias if bias is not None else x2d.new_empty(0), scale)
        else:
            y = F.linear(x2d, weight, bias)
            hip.ext().lora_add_nt_(y, t_u, bs)      # y += t_u @ bs^T
        ctx.save_for_backward(x2d, mask if mask is not None else x2d.new_empty(0),
                              t_u, weight, lora_A, bs)
        ctx.scale = scale
        ctx.dropout_p = dropout_p if use_dropout else 0.0
        ctx.has_bias = bias is not None
        return y.view(*in_shape[:-1], N)

    @staticmethod
    def backward(ctx, dy):
        x2d, mask, t_u, weight, lora_A, bs = ctx.saved_tensors
        scale, p = ctx.scale, ctx.dropout_p
        N = weight.shape[0]
        dy2d = dy.contiguous().view(-1, N)

        u_s = dy2d @ bs                             # [M, r] = s * dy @ B
        dx = dy2d @ weight                          # [M, K] main path
        hip.ext().lora_add_nn_(dx, u_s, lora_A,
                               mask if p > 0 else mask.new_empty(0, dtype=torch.uint8),
                               1.0 / (1.0 - p) if p > 0 else 1.0)

        # dA = u_s^T @ dropout(x), dB = s * dy^T @ t_u = (t_u^T @ dy)^T * s —
        # via the chunked-M skinny_grad kernel (hipBLASLt launches these
        # 32-WG wide); the dropout mask re-applies inline while staging x,
        # and scale/transpose/cast fold into the combine stage
        empty_mask = mask.new_empty(0, dtype=torch.uint8)
        dA = hip.ext().skinny_grad(u_s, x2d, mask if p > 0 else empty_mask,
                                   1.0 / (1.0 - p) if p > 0 else 1.0,
                                   1.0, False, lora_A.dtype)
        dB = hip.ext().skinny_grad(t_u, dy2d, empty_mask, 1.0, scale, True, bs.dtype)
        dw = dy2d.t() @ x2d if ctx.needs_input_grad[1] else None
        dbias = dy2d.sum(0) if ctx.has_bias and ctx.needs_input_grad[2] else None
        return (dx.view(dy.shape[:-1] + (weight.shape[1],)), dw, dbias,
                dA, dB, None, None, None)


class _QuantizedLoRALinear(torch.autograd.Function):
    """K15 path: frozen W stays NF4/int8-packed end to end.  Forward runs
    the dequant-fused MFMA GEMM (aligned NF4 shapes) or a transient
    materialize; backward re-dequantizes transiently for dX.  Unlike the
    dense path, NO dense [N,K] W is ever saved for backward — the HBM
    footprint of the frozen weights stays at the packed size (the entire
    point of --quantize; reference runs bnb matmul_4bit, relora.py:314-317).
    """

    @staticmethod
    def forward(ctx, x, bias, lora_A, lora_B, scale, dropout_p, training, qw):
        in_shape = x.shape
        K = in_shape[-1]
        N = qw.out_features
        x2d = x.contiguous().view(-1, K)
        M = x2d.shape[0]
        use_dropout = dropout_p > 0 and training
        if use_dropout:
            seed = _next_dropout_seed() & 0x7FFFFFFFFFFFFFFF
            xd, mask = hip.ext().dropout_mask_fwd(x2d, dropout_p, seed)
        else:
            xd, mask = x2d, None
        t_u = xd @ lora_A.t()
        del xd
        r = lora_A.shape[0]
        empty = x2d.new_empty(0)
        aligned = (M % 256 == 0 and N % 256 == 0 and r % 64 == 0
                   and r <= 256 and _FUSED_K1 != "0")
        fused_nf4 = aligned and qw.mode == "4bit" and K % 64 == 0
        fused_i8 = aligned and qw.mode == "8bit" and K % 256 == 0
        if fused_nf4:
            y = hip.ext().fused_nf4_gemm(
                x2d, qw.qdata, qw.absmax, N, t_u, lora_B,
                bias if bias is not None else empty, scale)
        elif fused_i8:
            y = hip.ext().fused_int8_gemm(
                x2d, qw.qdata, qw.absmax, N, t_u, lora_B,
                bias if bias is not None else empty, scale)
        else:
            w = qw.materialize(torch.bfloat16)  # transient; freed below
            y = F.linear(x2d, w, bias)
            del w
            hip.ext().lora_add_nt_(y, t_u, lora_B * scale)
        ctx.save_for_backward(x2d, mask if mask is not None else empty,
                              t_u, lora_A, lora_B)
        ctx.qw = qw
        ctx.scale = scale
        ctx.dropout_p = dropout_p if use_dropout else 0.0
        ctx.has_bias = bias is not None
        return y.view(*in_shape[:-1], N)

    @staticmethod
    def backward(ctx, dy):
        x2d, mask, t_u, lora_A, lora_B = ctx.saved_tensors
        scale, p = ctx.scale, ctx.dropout_p
        qw = ctx.qw
        N = qw.out_features
        dy2d = dy.contiguous().view(-1, N)
        bs = lora_B * scale
        u_s = dy2d @ bs
        w = qw.materialize(torch.bfloat16)  # transient dense W for dX only
        dx = dy2d @ w
        del w
        hip.ext().lora_add_nn_(dx, u_s, lora_A,
                               mask if p > 0 else mask.new_empty(0, dtype=torch.uint8),
                               1.0 / (1.0 - p) if p > 0 else 1.0)
        empty_mask = mask.new_empty(0, dtype=torch.uint8)
        dA = hip.ext().skinny_grad(u_s, x2d, mask if p > 0 else empty_mask,
                                   1.0 / (1.0 - p) if p > 0 else 1.0,
                                   1.0, False, lora_A.dtype)
        dB = hip.ext().skinny_grad(t_u, dy2d, empty_mask, 1.0, scale, True, lora_B.dtype)
        dbias = dy2d.sum(0) if ctx.has_bias else None
        return (dx.view(dy.shape[:-1] + (qw.in_features,)), dbias,
                dA, dB, None, None, None, None)


class _HipDropout(torch.autograd.Function):
    """Standalone dropout with the philox packed-bit mask kernels (1 bit per
    element vs torch's byte mask; the composed-path replacement for
    F.dropout on odd-dim linears)."""

    @staticmethod
    def forward(ctx, x, p):
        xc = x.contiguous()
        shape = xc.shape
        seed = _next_dropout_seed() & 0x7FFFFFFFFFFFFFFF
        xd, mask = hip.ext().dropout_mask_fwd(xc.view(-1, shape[-1]), p, seed)
        ctx.save_for_backward(mask)
        ctx.p = p
        return xd.view(shape)

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        dx = hip.ext().dropout_mask_bwd(dy.contiguous(), mask, ctx.p)
        return dx.view(dy.shape), None


def _fused_ok(x, weight, lora_A, scale, lora_only):
    if weight is None:
        return False
    r = lora_A.shape[0]
    # odd in/out dims (llama_1b intermediate 5461) are CORRECT through the
    # fused kernels (alignment-guarded fallbacks) but measured slower e2e
    # than the composed path even after its scale-folding trim (58.7k vs
    # 66.0k tok/s on the flagship, gpurun r2c A/B) — unaligned rows defeat
    # every vectorized RMW path.  RELORA_AMD_LORA_ODD=fused re-routes them
    # for future A/Bs.
    dims_ok = (x.shape[-1] % 8 == 0 and weight.shape[0] % 8 == 0
               or os.environ.get("RELORA_AMD_LORA_ODD", "torch") == "fused")
    return (hip.use_hip(x, "lora") and not lora_only and not torch.is_tensor(scale)
            and x.dtype == torch.bfloat16 and weight.dtype == torch.bfloat16
            and r % 32 == 0 and r <= 256 and dims_ok
            and os.environ.get("RELORA_AMD_LORA_PATH", "fused") != "torch")


def lora_linear(x, weight, bias, lora_A, lora_B, scale, dropout_p=0.0,
                training=False, lora_only=False, quantized_weight=None):
    """y = x W^T (+b) + s * dropout(x) A^T B^T.

    GPU bf16 path: _FusedLoRALinear (MFMA rank-r accumulate kernels).
    Otherwise: composed torch ops (CPU path and the numerics oracle).
    `scale` may be a python float or a 0-d tensor (trainable scaling,
    already passed through tanh by the caller) — tensor scale uses the
    composed path so autograd reaches it.
    """
    if quantized_weight is not None:
        if (hip.use_hip(x, "lora") and not lora_only and not torch.is_tensor(scale)
                and x.dtype == torch.bfloat16):
            return _QuantizedLoRALinear.apply(x, bias, lora_A, lora_B,
                                              float(scale), dropout_p, training,
                                              quantized_weight)
        # CPU / tensor-scale fallback: transient materialize + composed ops
        weight = quantized_weight.materialize(x.dtype if x.is_floating_point()
                                              else torch.float32)
    if _fused_ok(x, weight, lora_A, scale, lora_only):
        return _FusedLoRALinear.apply(x, weight, bias, lora_A, lora_B,
                                      float(scale), dropout_p, training)
    if dropout_p > 0 and training and x.is_cuda and x.dtype == torch.bfloat16 \
            and hip.use_hip(x, "dropout"):
        xd = _HipDropout.apply(x, dropout_p)  # packed 1-bit mask, philox
    elif dropout_p > 0:
        xd = F.dropout(x, p=dropout_p, training=training)
    else:
        xd = x
    if x.is_cuda:
        # fold the scale into the [M,r] intermediate (45x smaller than the
        # [M,N] product the naive composition scales) — same math, removes
        # a full-size elementwise pass; tensor scale keeps autograd
        lora_out = F.linear(F.linear(xd, lora_A) * scale, lora_B)
        if lora_only:
            return lora_out
        return F.linear(x, weight, bias).add_(lora_out)
    # CPU path stays the bit-exact reference composition (parity oracle)
    lora_out = F.linear(F.linear(xd, lora_A), lora_B)
    if lora_only:
        return lora_out * scale
    return F.linear(x, weight, bias) + lora_out * scale
