"""Model-family tests: shapes, loss-goes-down on a tiny synthetic task,
gradient checkpointing equivalence, pythia parallel residual, KV-cache
decode consistency."""

import torch

from relora_amd.models.llama import LlamaForCausalLM
from relora_amd.models.pythia import GPTNeoXForCausalLM


def test_llama_forward_shapes(tiny_llama_config):
    model = LlamaForCausalLM(tiny_llama_config)
    x = torch.randint(0, tiny_llama_config.vocab_size, (2, 24))
    out = model(input_ids=x)
    assert out.logits.shape == (2, 24, tiny_llama_config.vocab_size)
    out = model(input_ids=x, labels=x)
    assert out.loss.dim() == 0


def test_pythia_forward_shapes(tiny_pythia_config):
    model = GPTNeoXForCausalLM(tiny_pythia_config)
    x = torch.randint(0, tiny_pythia_config.vocab_size, (2, 24))
    out = model(input_ids=x, labels=x)
    assert out.loss.dim() == 0


def _overfit(model, vocab, steps=60):
    torch.manual_seed(0)
    x = torch.randint(0, vocab, (4, 32))
    opt = torch.optim.AdamW(model.parameters(), lr=3e-3)
    first = None
    for _ in range(steps):
        loss = model(input_ids=x, labels=x).loss
        if first is None:
            first = loss.item()
        opt.zero_grad()
        loss.backward()
        opt.step()
    return first, loss.item()


def test_llama_loss_goes_down(tiny_llama_config):
    model = LlamaForCausalLM(tiny_llama_config)
    first, last = _overfit(model, tiny_llama_config.vocab_size)
    assert last < first * 0.7, (first, last)


def test_pythia_loss_goes_down(tiny_pythia_config):
    model = GPTNeoXForCausalLM(tiny_pythia_config)
    first, last = _overfit(model, tiny_pythia_config.vocab_size)
    assert last < first * 0.7, (first, last)


def test_llama_gradient_checkpointing_equivalence(tiny_llama_config):
    torch.manual_seed(0)
    model = LlamaForCausalLM(tiny_llama_config)
    x = torch.randint(0, tiny_llama_config.vocab_size, (2, 16))

    loss1 = model(input_ids=x, labels=x).loss
    loss1.backward()
    grads1 = {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}
    model.zero_grad()

    model.model.gradient_checkpointing = True
    model.train()
    loss2 = model(input_ids=x, labels=x).loss
    loss2.backward()
    assert torch.allclose(loss1, loss2, atol=1e-6)
    for n, p in model.named_parameters():
        if p.grad is not None and n in grads1:
            assert torch.allclose(grads1[n], p.grad, atol=1e-5), n


def test_pythia_parallel_residual_flag(tiny_pythia_config):
    torch.manual_seed(0)
    m_par = GPTNeoXForCausalLM(tiny_pythia_config)
    cfg2 = type(tiny_pythia_config)(**{**tiny_pythia_config.to_dict(),
                                       "use_parallel_residual": False})
    torch.manual_seed(0)
    m_seq = GPTNeoXForCausalLM(cfg2)
    x = torch.randint(0, 256, (1, 8))
    with torch.no_grad():
        a = m_par(input_ids=x).logits
        b = m_seq(input_ids=x).logits
    assert not torch.allclose(a, b)


def test_llama_kv_cache_decode(tiny_llama_config):
    torch.manual_seed(0)
    model = LlamaForCausalLM(tiny_llama_config).eval()
    model.fused_ce = False
    x = torch.randint(0, tiny_llama_config.vocab_size, (1, 12))
    with torch.no_grad():
        full = model(input_ids=x).logits
        # incremental: prefill 11, decode 1
        out = model(input_ids=x[:, :11], use_cache=True)
        step = model(input_ids=x[:, 11:], past_key_values=out.past_key_values, use_cache=True)
    assert torch.allclose(full[:, -1], step.logits[:, -1], atol=1e-4)


def test_save_load_roundtrip_plain(tiny_llama_config, tmp_path):
    from relora_amd.models import build_model_from_config, load_model_config
    from relora_amd.utils.checkpoint import load_state_dict_compat, save_pretrained_compat

    torch.manual_seed(0)
    model = LlamaForCausalLM(tiny_llama_config)
    save_pretrained_compat(model, tmp_path / "m")
    cfg = load_model_config(str(tmp_path / "m"))
    model2 = build_model_from_config(cfg)
    model2.load_state_dict(load_state_dict_compat(str(tmp_path / "m")), strict=True)
    x = torch.randint(0, tiny_llama_config.vocab_size, (1, 8))
    with torch.no_grad():
        assert torch.allclose(
            model(input_ids=x, labels=x).loss, model2(input_ids=x, labels=x).loss
        )


def test_rope_cache_stays_fp32_after_model_cast():
    """A model-wide .to(bf16) casts registered buffers; the rotary modules
    must still hand fp32 tables to the RoPE kernel (its hard contract —
    this regression aborted the first GPU smoke run)."""
    from relora_amd.models import build_model_from_config, load_model_config

    for cfg_path in ("configs/llama_9m.json",):
        cfg = load_model_config(cfg_path)
        model = build_model_from_config(cfg).to(dtype=torch.bfloat16)
        x = torch.randint(0, cfg.vocab_size, (1, 8))
        out = model(input_ids=x, labels=x)
        assert torch.isfinite(out.loss.float())
        rot = model.model.layers[0].self_attn.rotary_emb
        cos, sin = rot(torch.zeros(1, dtype=torch.bfloat16), seq_len=8)
        assert cos.dtype == torch.float32 and sin.dtype == torch.float32

    from relora_amd.models.pythia import GPTNeoXConfig, GPTNeoXForCausalLM

    pcfg = GPTNeoXConfig(vocab_size=128, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64,
                         max_position_embeddings=32)
    pm = GPTNeoXForCausalLM(pcfg).to(dtype=torch.bfloat16)
    x = torch.randint(0, 128, (1, 8))
    out = pm(input_ids=x, labels=x)
    assert torch.isfinite(out.loss.float())
    rot = pm.gpt_neox.layers[0].attention.rotary_emb
    cos, sin = rot(torch.zeros(1, dtype=torch.bfloat16), seq_len=8)
    assert cos.dtype == torch.float32 and sin.dtype == torch.float32


def test_gradient_checkpointing_matches_plain():
    """Grad checkpointing produces the same loss and gradients."""
    from relora_amd.models import build_model_from_config, load_model_config

    torch.manual_seed(0)
    cfg = load_model_config("configs/llama_9m.json")
    model = build_model_from_config(cfg)
    x = torch.randint(0, cfg.vocab_size, (2, 32))

    loss_plain = model(input_ids=x, labels=x).loss
    loss_plain.backward()
    grads_plain = {n: p.grad.clone() for n, p in model.named_parameters()
                   if p.grad is not None}
    model.zero_grad()

    model.gradient_checkpointing_enable()
    loss_ckpt = model(input_ids=x, labels=x).loss
    loss_ckpt.backward()
    assert torch.allclose(loss_plain, loss_ckpt, atol=1e-6)
    for n, p in model.named_parameters():
        if p.grad is not None:
            assert torch.allclose(grads_plain[n], p.grad, atol=1e-5), n


def test_kv_cache_incremental_decode_matches_full():
    """use_cache incremental decode equals the full forward logits
    (the reference's generation affordance, modeling_llama.py past_key_values)."""
    from relora_amd.models import build_model_from_config, load_model_config

    torch.manual_seed(1)
    cfg = load_model_config("configs/llama_9m.json")
    model = build_model_from_config(cfg).eval()
    x = torch.randint(0, cfg.vocab_size, (1, 12))
    with torch.no_grad():
        full = model(input_ids=x).logits
        # prefill 8 tokens, then decode 4 one by one
        out = model(input_ids=x[:, :8], use_cache=True)
        past = out.past_key_values
        logits = [out.logits]
        for t in range(8, 12):
            step = model(input_ids=x[:, t:t + 1], past_key_values=past, use_cache=True)
            past = step.past_key_values
            logits.append(step.logits)
        inc = torch.cat(logits, dim=1)
    assert torch.allclose(full, inc, atol=2e-4, rtol=1e-4), (full - inc).abs().max()


def test_pythia_rope_scaling_variants():
    """linear and dynamic-NTK RoPE scaling paths run and differ from base."""
    from relora_amd.models.pythia import GPTNeoXConfig, GPTNeoXForCausalLM

    torch.manual_seed(2)
    base_cfg = dict(vocab_size=128, hidden_size=64, num_hidden_layers=1,
                    num_attention_heads=2, intermediate_size=128,
                    max_position_embeddings=16, rotary_pct=0.5)
    x = torch.randint(0, 128, (1, 32))  # beyond max_position -> scaling engages
    outs = {}
    for label, scaling in (("base", None),
                           ("linear", {"type": "linear", "factor": 4.0}),
                           ("dynamic", {"type": "dynamic", "factor": 4.0})):
        torch.manual_seed(3)
        m = GPTNeoXForCausalLM(GPTNeoXConfig(**base_cfg, rope_scaling=scaling)).eval()
        with torch.no_grad():
            outs[label] = m(input_ids=x).logits
    assert not torch.allclose(outs["base"], outs["linear"])
    assert not torch.allclose(outs["base"], outs["dynamic"])


def test_pythia_tied_embeddings():
    """tie_word_embeddings=True shares embed_in/embed_out storage (the
    pythia tied-head variant; reference modeling_pythia.py head wiring),
    gradients flow, and the untied default does not share."""
    import torch as _t

    from relora_amd.models.pythia import GPTNeoXConfig, GPTNeoXForCausalLM

    base = dict(vocab_size=128, hidden_size=32, num_hidden_layers=2,
                num_attention_heads=4, intermediate_size=64,
                max_position_embeddings=64, rotary_pct=0.25)
    tied = GPTNeoXForCausalLM(GPTNeoXConfig(**base, tie_word_embeddings=True))
    assert (tied.embed_out.weight.data_ptr()
            == tied.gpt_neox.embed_in.weight.data_ptr())
    untied = GPTNeoXForCausalLM(GPTNeoXConfig(**base, tie_word_embeddings=False))
    assert (untied.embed_out.weight.data_ptr()
            != untied.gpt_neox.embed_in.weight.data_ptr())

    x = _t.randint(0, 128, (2, 16))
    loss = tied(input_ids=x, labels=x).loss
    loss.backward()
    assert tied.gpt_neox.embed_in.weight.grad is not None
    # save/load keeps the tie
    sd = tied.state_dict()
    reloaded = GPTNeoXForCausalLM(GPTNeoXConfig(**base, tie_word_embeddings=True))
    reloaded.load_state_dict(sd)
    assert (reloaded.embed_out.weight.data_ptr()
            == reloaded.gpt_neox.embed_in.weight.data_ptr())


def test_llama_tie_word_embeddings_honored():
    """tie_word_embeddings=True shares lm_head/embed_tokens storage; the
    default (and every shipped configs/llama_*.json) stays untied."""
    import torch as _t

    from relora_amd.models.config import LlamaConfig
    from relora_amd.models.llama import LlamaForCausalLM

    base = dict(vocab_size=128, hidden_size=32, intermediate_size=64,
                num_hidden_layers=2, num_attention_heads=4,
                max_position_embeddings=64)
    for tie in (True, False):
        m = LlamaForCausalLM(LlamaConfig(**base, tie_word_embeddings=tie))
        shared = (m.lm_head.weight.data_ptr()
                  == m.model.embed_tokens.weight.data_ptr())
        assert shared == tie
        x = _t.randint(0, 128, (1, 8))
        m(input_ids=x, labels=x).loss.backward()


def test_generate_matches_manual_greedy():
    """model.generate (GenerationMixin + DynamicCache under the installed
    transformers) produces exactly the tokens of a manual greedy
    full-recompute loop, for both model families."""
    import torch as _t

    from relora_amd.models.config import GPTNeoXConfig, LlamaConfig
    from relora_amd.models.llama import LlamaForCausalLM
    from relora_amd.models.pythia import GPTNeoXForCausalLM

    _t.manual_seed(0)
    x = _t.randint(2, 128, (1, 5))
    models = [
        LlamaForCausalLM(LlamaConfig(
            vocab_size=128, hidden_size=32, intermediate_size=64,
            num_hidden_layers=2, num_attention_heads=4,
            max_position_embeddings=64, bos_token_id=0, eos_token_id=1)),
        GPTNeoXForCausalLM(GPTNeoXConfig(
            vocab_size=128, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=4, intermediate_size=64,
            max_position_embeddings=64, rotary_pct=0.25,
            bos_token_id=0, eos_token_id=1)),
    ]
    for m in models:
        m.eval()
        out = m.generate(x, max_new_tokens=6, do_sample=False)
        cur = x.clone()
        for _ in range(6):
            with _t.no_grad():
                cur = _t.cat([cur, m(input_ids=cur).logits[:, -1:].argmax(-1)],
                             dim=1)
        assert _t.equal(out, cur), type(m).__name__


def test_hf_from_pretrained_classmethod_exact():
    """PreTrainedModel.from_pretrained on our classes is bit-exact.
    Regression: transformers materializes models from the META device, which
    voids non-persistent buffers — the RoPE caches must detect that and
    rebuild (they silently held garbage before; logits were off by ~1e-2)."""
    import tempfile

    import torch as _t

    from relora_amd.models.config import GPTNeoXConfig, LlamaConfig
    from relora_amd.models.llama import LlamaForCausalLM
    from relora_amd.models.pythia import GPTNeoXForCausalLM
    from relora_amd.utils.checkpoint import save_pretrained_compat

    _t.manual_seed(0)
    x = _t.randint(0, 128, (1, 8))
    cases = [
        (LlamaForCausalLM, LlamaConfig(
            vocab_size=128, hidden_size=32, intermediate_size=64,
            num_hidden_layers=2, num_attention_heads=4,
            max_position_embeddings=64)),
        (GPTNeoXForCausalLM, GPTNeoXConfig(
            vocab_size=128, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=4, intermediate_size=64,
            max_position_embeddings=64, rotary_pct=0.25)),
    ]
    for cls, cfg in cases:
        m = cls(cfg).eval()
        d = tempfile.mkdtemp()
        save_pretrained_compat(m, d)
        m2 = cls.from_pretrained(d).eval()
        with _t.no_grad():
            a = m(input_ids=x).logits
            b = m2(input_ids=x).logits
        assert _t.equal(a, b), (cls.__name__, (a - b).abs().max())


def test_rope_cache_rebuilds_on_cast_after_forward():
    """The forward-then-cast order: a model that already ran (validity flag
    set) and is THEN cast to bf16 must still rebuild its fp32 RoPE tables
    on the next forward (the kernel contract)."""
    import torch as _t

    from relora_amd.models.config import LlamaConfig
    from relora_amd.models.llama import LlamaForCausalLM

    m = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=4,
        max_position_embeddings=64)).eval()
    x = _t.randint(0, 128, (1, 8))
    with _t.no_grad():
        m(input_ids=x)                      # sets the validity flag
    m = m.to(_t.bfloat16)                   # casts the cached tables
    with _t.no_grad():
        m(input_ids=x)
    rot = m.model.layers[0].self_attn.rotary_emb
    assert rot.cos_cached.dtype == _t.float32


def test_cache_compat_unit():
    """cache_compat: legacy passthrough, empty-cache handling, and cache
    reconstruction in the template's type."""
    import torch as _t
    from transformers.cache_utils import DynamicCache

    from relora_amd.models.cache_compat import cache_like, cache_to_legacy

    assert cache_to_legacy(None) is None
    assert cache_to_legacy(()) is None
    legacy = ((_t.zeros(1, 2, 3, 4), _t.ones(1, 2, 3, 4)),)
    assert cache_to_legacy(legacy) is legacy
    assert cache_to_legacy(DynamicCache()) is None  # fresh/empty

    filled = DynamicCache()
    filled.update(_t.zeros(1, 2, 3, 4), _t.ones(1, 2, 3, 4), 0)
    back = cache_to_legacy(filled)
    assert isinstance(back, tuple) and back[0][0].shape == (1, 2, 3, 4)

    # packaging follows the template type
    assert isinstance(cache_like(list(legacy), None), tuple)
    rebuilt = cache_like(list(legacy), DynamicCache())
    assert isinstance(rebuilt, DynamicCache)
    assert rebuilt.get_seq_length() == 3
    assert cache_like([], DynamicCache()) is None


def test_llama_deferred_residual_matches_materialized():
    """The cross-layer residual deferral (defer_add) must be numerically
    identical to the materialized path (output_hidden_states=True disables
    the deferral), and hidden-states output must still be per-layer sums."""
    import torch

    from relora_amd.models import build_model_from_config, load_model_config

    torch.manual_seed(0)
    cfg = load_model_config("configs/llama_9m.json")
    model = build_model_from_config(cfg).eval()
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    with torch.no_grad():
        out_deferred = model.model(input_ids=x)  # defer path (no hidden states)
        out_material = model.model(input_ids=x, output_hidden_states=True)
    assert torch.equal(out_deferred.last_hidden_state,
                       out_material.last_hidden_state)
    assert len(out_material.hidden_states) == cfg.num_hidden_layers + 1
