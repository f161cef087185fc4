"""CPU oracle tests for the functional op layer: each op's torch path is
checked against an independent composition (the same contract the HIP
kernels are tested against on GPU in test_gpu_kernels.py)."""

import pytest
import torch
import torch.nn.functional as F

from relora_amd import ops


def test_rmsnorm_matches_reference_formula():
    torch.manual_seed(0)
    x = torch.randn(4, 32, dtype=torch.float32)
    w = torch.randn(32)
    y = ops.rmsnorm(x, w, eps=1e-6)
    var = x.pow(2).mean(-1, keepdim=True)
    expected = w * (x * torch.rsqrt(var + 1e-6))
    assert torch.allclose(y, expected, atol=1e-6)


def test_rmsnorm_bf16_dtype_semantics():
    torch.manual_seed(0)
    x = torch.randn(4, 32).bfloat16()
    w = torch.randn(32).bfloat16()
    y = ops.rmsnorm(x, w, eps=1e-6)
    assert y.dtype == torch.bfloat16
    # fp32 variance: matches fp32 computation within bf16 rounding
    ref = (w.float() * (x.float() * torch.rsqrt(x.float().pow(2).mean(-1, keepdim=True) + 1e-6)))
    assert torch.allclose(y.float(), ref, atol=3e-2)


def test_rope_matches_rotate_half():
    torch.manual_seed(0)
    B, nh, S, hd = 2, 4, 16, 8
    q = torch.randn(B, nh, S, hd)
    k = torch.randn(B, nh, S, hd)
    cos, sin = ops.build_rope_cache(hd, S)
    qo, ko = ops.rope(q, k, cos, sin)
    qe = q * cos[:S] + ops.rotate_half(q) * sin[:S]
    ke = k * cos[:S] + ops.rotate_half(k) * sin[:S]
    assert torch.allclose(qo, qe, atol=1e-6)
    assert torch.allclose(ko, ke, atol=1e-6)


def test_rope_partial_rotary():
    B, nh, S, hd, rot = 1, 2, 8, 16, 4
    q = torch.randn(B, nh, S, hd)
    k = torch.randn(B, nh, S, hd)
    cos, sin = ops.build_rope_cache(rot, S)
    qo, ko = ops.rope(q, k, cos, sin)
    # pass-through part untouched
    assert torch.equal(qo[..., rot:], q[..., rot:])
    assert torch.equal(ko[..., rot:], k[..., rot:])
    assert not torch.allclose(qo[..., :rot], q[..., :rot])


def test_rope_inverse():
    """RoPE at position 0 is identity (cos=1, sin=0)."""
    q = torch.randn(1, 1, 1, 8)
    cos, sin = ops.build_rope_cache(8, 4)
    qo, _ = ops.rope(q, q.clone(), cos, sin)
    assert torch.allclose(qo, q, atol=1e-6)


def test_swiglu():
    g = torch.randn(16, 8, requires_grad=True)
    u = torch.randn(16, 8, requires_grad=True)
    y = ops.swiglu(g, u)
    assert torch.allclose(y, F.silu(g) * u)
    y.sum().backward()
    assert g.grad is not None and u.grad is not None


def test_flash_attention_cpu_matches_naive():
    torch.manual_seed(0)
    B, nh, S, hd = 2, 2, 16, 8
    q, k, v = (torch.randn(B, nh, S, hd) for _ in range(3))
    out = ops.flash_attention(q, k, v, causal=True)
    scores = (q @ k.transpose(-1, -2)) / hd ** 0.5
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool), diagonal=1)
    scores = scores.masked_fill(mask, float("-inf"))
    expected = torch.softmax(scores, dim=-1) @ v
    assert torch.allclose(out, expected, atol=1e-5)


# ---------------------------------------------------------------------------
# fused chunked cross-entropy
# ---------------------------------------------------------------------------


@pytest.mark.parametrize("chunk", [7, 64, 10_000])
def test_fused_ce_matches_torch(monkeypatch, chunk):
    import relora_amd.ops.functional as fops

    monkeypatch.setattr(fops, "_CE_CHUNK", chunk)
    torch.manual_seed(0)
    M, H, V = 64, 32, 101
    hidden = torch.randn(M, H, requires_grad=True)
    weight = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (M,))
    labels[5] = -100
    labels[17] = -100

    loss = ops.fused_cross_entropy(hidden, weight, labels)
    loss.backward()

    h2 = hidden.detach().clone().requires_grad_(True)
    w2 = weight.detach().clone().requires_grad_(True)
    ref = F.cross_entropy(h2 @ w2.t(), labels)
    ref.backward()

    assert torch.allclose(loss, ref, atol=1e-5)
    assert torch.allclose(hidden.grad, h2.grad, atol=1e-5)
    assert torch.allclose(weight.grad, w2.grad, atol=1e-4)


def test_fused_ce_bf16():
    torch.manual_seed(0)
    M, H, V = 32, 16, 50
    hidden = torch.randn(M, H).bfloat16().requires_grad_(True)
    weight = torch.randn(V, H).bfloat16().requires_grad_(True)
    labels = torch.randint(0, V, (M,))
    loss = ops.fused_cross_entropy(hidden, weight, labels)
    loss.backward()
    ref = F.cross_entropy((hidden.detach().float() @ weight.detach().float().t()), labels)
    assert torch.allclose(loss.float(), ref, atol=3e-2)


def test_fused_ce_in_model_matches_unfused(tiny_llama_config):
    from relora_amd.models.llama import LlamaForCausalLM

    torch.manual_seed(0)
    model = LlamaForCausalLM(tiny_llama_config)
    x = torch.randint(0, tiny_llama_config.vocab_size, (2, 24))
    loss_fused = model(input_ids=x, labels=x).loss
    model.fused_ce = False
    loss_unfused = model(input_ids=x, labels=x).loss
    assert torch.allclose(loss_fused, loss_unfused, atol=1e-5)


# ---------------------------------------------------------------------------
# lora_linear
# ---------------------------------------------------------------------------


def test_lora_linear_composition():
    torch.manual_seed(0)
    M, K, N, r = 8, 16, 12, 4
    x = torch.randn(M, K, requires_grad=True)
    W = torch.randn(N, K)
    A = torch.randn(r, K, requires_grad=True)
    B = torch.randn(N, r, requires_grad=True)
    s = 2.0
    y = ops.lora_linear(x, W, None, A, B, s, dropout_p=0.0, training=True)
    expected = x @ W.t() + (x @ A.t() @ B.t()) * s
    assert torch.allclose(y, expected, atol=1e-5)
    y.sum().backward()
    assert x.grad is not None and A.grad is not None and B.grad is not None


def test_lora_linear_lora_only():
    x = torch.randn(4, 8)
    A = torch.randn(2, 8)
    B = torch.randn(6, 2)
    y = ops.lora_linear(x, None, None, A, B, 0.5, lora_only=True)
    assert torch.allclose(y, (x @ A.t() @ B.t()) * 0.5, atol=1e-6)


def test_fused_ce_save_logits_flag(monkeypatch):
    """RELORA_AMD_CE_SAVE_LOGITS=1 must produce identical loss and grads
    (it only skips the backward logits recompute)."""
    import importlib

    import relora_amd.ops.functional as fn

    torch.manual_seed(0)
    M, H, V = 64, 32, 97
    hidden = torch.randn(M, H, requires_grad=True)
    weight = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (M,))
    labels[::7] = -100

    loss_a = fn.fused_cross_entropy(hidden, weight, labels)
    loss_a.backward()
    ga, gw = hidden.grad.clone(), weight.grad.clone()
    hidden.grad = weight.grad = None

    monkeypatch.setenv("RELORA_AMD_CE_SAVE_LOGITS", "1")
    importlib.reload(fn)
    try:
        loss_b = fn.fused_cross_entropy(hidden, weight, labels)
        loss_b.backward()
        assert torch.allclose(loss_a, loss_b, atol=1e-6)
        assert torch.allclose(ga, hidden.grad, atol=1e-6)
        assert torch.allclose(gw, weight.grad, atol=1e-6)
    finally:
        monkeypatch.delenv("RELORA_AMD_CE_SAVE_LOGITS")
        importlib.reload(fn)
