"""Model configuration classes.

Self-contained HF-compatible configs for the two model families the engine
supports (parity with reference `peft_pretraining/modeling_llama.py:33` which
uses transformers' LlamaConfig, and `modeling_pythia.py` / GPTNeoXConfig).

We keep our own classes (subclassing `transformers.PretrainedConfig`) so the
on-disk JSON contract (`configs/llama_*.json`, fields like
`max_sequence_length`) is stable regardless of the installed transformers
version.
"""

from transformers import PretrainedConfig


class LlamaConfig(PretrainedConfig):
    model_type = "llama"

    def __init__(
        self,
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=11008,
        num_hidden_layers=32,
        num_attention_heads=32,
        hidden_act="silu",
        max_position_embeddings=2048,
        initializer_range=0.02,
        rms_norm_eps=1e-6,
        use_cache=True,
        pad_token_id=-1,
        bos_token_id=0,
        eos_token_id=1,
        tie_word_embeddings=False,
        rope_theta=10000.0,
        **kwargs,
    ):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.hidden_act = hidden_act
        # the reference config JSONs carry `max_sequence_length`; the reference
        # model consumes `max_position_embeddings` (default 2048) for the RoPE
        # cache, which grows lazily anyway.
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.rms_norm_eps = rms_norm_eps
        self.use_cache = use_cache
        self.rope_theta = rope_theta
        super().__init__(
            pad_token_id=pad_token_id,
            bos_token_id=bos_token_id,
            eos_token_id=eos_token_id,
            tie_word_embeddings=tie_word_embeddings,
            **kwargs,
        )


class GPTNeoXConfig(PretrainedConfig):
    model_type = "gpt_neox"

    def __init__(
        self,
        vocab_size=50432,
        hidden_size=6144,
        num_hidden_layers=44,
        num_attention_heads=64,
        intermediate_size=24576,
        hidden_act="gelu",
        rotary_pct=0.25,
        rotary_emb_base=10000,
        attention_dropout=0.0,
        hidden_dropout=0.0,
        classifier_dropout=0.1,
        max_position_embeddings=2048,
        initializer_range=0.02,
        layer_norm_eps=1e-5,
        use_cache=True,
        bos_token_id=0,
        eos_token_id=2,
        tie_word_embeddings=False,
        use_parallel_residual=True,
        rope_scaling=None,
        **kwargs,
    ):
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.rotary_pct = rotary_pct
        self.rotary_emb_base = rotary_emb_base
        self.attention_dropout = attention_dropout
        self.hidden_dropout = hidden_dropout
        self.classifier_dropout = classifier_dropout
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.use_cache = use_cache
        self.use_parallel_residual = use_parallel_residual
        self.rope_scaling = rope_scaling
        super().__init__(
            bos_token_id=bos_token_id,
            eos_token_id=eos_token_id,
            tie_word_embeddings=tie_word_embeddings,
            **kwargs,
        )


def load_model_config(path):
    """Load a model config JSON (LlamaConfig or GPTNeoXConfig by model_type)."""
    import json
    import os

    cfg_file = path
    if os.path.isdir(path):
        cfg_file = os.path.join(path, "config.json")
    with open(cfg_file) as f:
        d = json.load(f)
    model_type = d.get("model_type", "llama")
    if model_type == "llama":
        return LlamaConfig(**d)
    if model_type == "gpt_neox":
        return GPTNeoXConfig(**d)
    raise ValueError(f"Unknown model_type {model_type!r} in {cfg_file}")
