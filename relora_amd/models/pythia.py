"""GPTNeoX/Pythia decoder for MI355X.

Functional parity with the reference (`peft_pretraining/modeling_pythia.py`):
fused `query_key_value` projection with per-head [q|k|v] layout (:176-185),
partial rotary (`rotary_pct`, :186-197), linear / dynamic-NTK RoPE scaling
(:333-375), LayerNorm with bias, parallel residual (:443-448), GELU MLP
(:395-406), untied `embed_out` head (:701-812). Fresh implementation; hot
ops route through `relora_amd.ops` (HIP layernorm, RoPE, causal flash
attention) on gfx950.
"""

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F
import torch.utils.checkpoint
from transformers.modeling_outputs import BaseModelOutputWithPast, CausalLMOutputWithPast
from transformers.generation import GenerationMixin
from transformers.modeling_utils import PreTrainedModel

from relora_amd import ops
from relora_amd.models.config import GPTNeoXConfig


class GPTNeoXRotary(nn.Module):
    """cos/sin cache with optional linear / dynamic-NTK scaling."""

    def __init__(self, dim, max_position_embeddings, base=10000, scaling=None):
        super().__init__()
        self.dim = dim
        self.base = base
        self.max_position_embeddings = max_position_embeddings
        # transformers' PretrainedConfig normalizes rope_scaling to
        # {'rope_type': 'default', ...}; treat 'default' as no scaling
        stype = None
        sfactor = 1.0
        if scaling:
            stype = scaling.get("type", scaling.get("rope_type"))
            if stype == "default":
                stype = None
            sfactor = scaling.get("factor", 1.0)
        self.scaling_type = stype
        self.scaling_factor = sfactor
        self.max_seq_len_cached = 0
        self._build(max_position_embeddings, None)

    def _build(self, seq_len, device):
        base = self.base
        if self.scaling_type == "dynamic" and seq_len > self.max_position_embeddings:
            base = self.base * (
                (self.scaling_factor * seq_len / self.max_position_embeddings)
                - (self.scaling_factor - 1)
            ) ** (self.dim / (self.dim - 2))
        inv_freq = 1.0 / (
            base ** (torch.arange(0, self.dim, 2, dtype=torch.float32, device=device) / self.dim)
        )
        t = torch.arange(seq_len, dtype=torch.float32, device=device)
        if self.scaling_type == "linear":
            t = t / self.scaling_factor
        freqs = torch.outer(t, inv_freq)
        emb = torch.cat((freqs, freqs), dim=-1)
        self.register_buffer("cos_cached", emb.cos(), persistent=False)
        self.register_buffer("sin_cached", emb.sin(), persistent=False)
        self.max_seq_len_cached = seq_len

    def _cache_is_valid(self):
        # from_pretrained materializes from the meta device, leaving
        # non-persistent buffers uninitialized; cos(position 0) == 1 in any
        # real cache — verify once per process.  (The fp32-dtype check stays
        # SEPARATE in forward: a .to(dtype) can cast the cache later.)
        if getattr(self, "_cache_checked", False):
            return True
        ok = bool((self.cos_cached[0] == 1).all())
        self._cache_checked = ok
        return ok

    def forward(self, x, seq_len):
        if seq_len > self.max_seq_len_cached or (
            self.scaling_type == "dynamic" and seq_len != self.max_seq_len_cached
        ) or self.cos_cached.dtype != torch.float32 or not self._cache_is_valid():
            # fp32 tables are part of the RoPE kernel contract; a model-wide
            # .to(dtype) may have cast the buffers, and meta-device loading
            # may have voided them.
            self._build(max(seq_len, self.max_position_embeddings), x.device)
            self._cache_checked = True
        return self.cos_cached.to(x.device), self.sin_cached.to(x.device)


class GPTNeoXAttention(nn.Module):
    def __init__(self, config: GPTNeoXConfig):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.hidden_size = config.hidden_size
        if self.hidden_size % self.num_heads != 0:
            raise ValueError("hidden_size must be divisible by num_attention_heads")
        self.head_size = self.hidden_size // self.num_heads
        self.rotary_ndims = int(self.head_size * config.rotary_pct)
        self.rotary_emb = GPTNeoXRotary(
            self.rotary_ndims, config.max_position_embeddings,
            base=config.rotary_emb_base, scaling=config.rope_scaling,
        )
        self.query_key_value = nn.Linear(config.hidden_size, 3 * config.hidden_size)
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.attention_dropout_p = float(config.attention_dropout)

    def forward(self, hidden_states, position_ids=None, layer_past=None, use_cache=False):
        B, S, _ = hidden_states.shape
        qkv = self.query_key_value(hidden_states)
        # per-head [q|k|v] packing (reference modeling_pythia.py:176-185)
        qkv = qkv.view(B, S, self.num_heads, 3 * self.head_size)
        q = qkv[..., : self.head_size].permute(0, 2, 1, 3)
        k = qkv[..., self.head_size : 2 * self.head_size].permute(0, 2, 1, 3)
        v = qkv[..., 2 * self.head_size :].permute(0, 2, 1, 3)

        kv_seq_len = S
        if layer_past is not None:
            kv_seq_len += layer_past[0].shape[-2]
        cos, sin = self.rotary_emb(v, seq_len=kv_seq_len)
        if layer_past is not None and position_ids is None:
            position_ids = torch.arange(
                kv_seq_len - S, kv_seq_len, device=hidden_states.device
            ).unsqueeze(0)
        q, k = ops.rope(q, k, cos, sin, position_ids=position_ids)

        if layer_past is not None:
            k = torch.cat((layer_past[0], k), dim=-2)
            v = torch.cat((layer_past[1], v), dim=-2)
        present = (k, v) if use_cache else None

        dropout_p = self.attention_dropout_p if self.training else 0.0
        causal = q.shape[-2] > 1
        attn = ops.flash_attention(q, k, v, causal=causal, dropout_p=dropout_p)
        attn = attn.permute(0, 2, 1, 3).reshape(B, S, self.hidden_size)
        return self.dense(attn), present


class GPTNeoXMLP(nn.Module):
    def __init__(self, config: GPTNeoXConfig):
        super().__init__()
        self.dense_h_to_4h = nn.Linear(config.hidden_size, config.intermediate_size)
        self.dense_4h_to_h = nn.Linear(config.intermediate_size, config.hidden_size)

    def forward(self, hidden_states):
        hidden_states = self.dense_h_to_4h(hidden_states)
        hidden_states = ops.gelu(hidden_states)  # K8 HIP kernel on GPU
        return self.dense_4h_to_h(hidden_states)


class GPTNeoXLayer(nn.Module):
    def __init__(self, config: GPTNeoXConfig):
        super().__init__()
        self.use_parallel_residual = config.use_parallel_residual
        self.input_layernorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_layernorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.post_attention_dropout = nn.Dropout(config.hidden_dropout)
        self.post_mlp_dropout = nn.Dropout(config.hidden_dropout)
        self.attention = GPTNeoXAttention(config)
        self.mlp = GPTNeoXMLP(config)

    def _ln(self, ln, x):
        return ops.layernorm(x, ln.weight, ln.bias, ln.eps)

    def forward(self, hidden_states, position_ids=None, layer_past=None, use_cache=False):
        attn_out, present = self.attention(
            self._ln(self.input_layernorm, hidden_states),
            position_ids=position_ids, layer_past=layer_past, use_cache=use_cache,
        )
        attn_out = self.post_attention_dropout(attn_out)

        if self.use_parallel_residual:
            # x + attn(ln1(x)) + mlp(ln2(x))   (reference modeling_pythia.py:443-448)
            mlp_out = self.mlp(self._ln(self.post_attention_layernorm, hidden_states))
            mlp_out = self.post_mlp_dropout(mlp_out)
            hidden_states = mlp_out + attn_out + hidden_states
        else:
            attn_out = attn_out + hidden_states
            mlp_out = self.mlp(self._ln(self.post_attention_layernorm, attn_out))
            mlp_out = self.post_mlp_dropout(mlp_out)
            hidden_states = mlp_out + attn_out
        return hidden_states, present


class GPTNeoXPreTrainedModel(PreTrainedModel):
    config_class = GPTNeoXConfig
    base_model_prefix = "gpt_neox"
    supports_gradient_checkpointing = True
    _no_split_modules = ["GPTNeoXLayer"]

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.padding_idx is not None:
                module.weight.data[module.padding_idx].zero_()
        elif isinstance(module, nn.LayerNorm):
            module.bias.data.zero_()
            module.weight.data.fill_(1.0)

    def _set_gradient_checkpointing(self, module, value=False):
        if isinstance(module, GPTNeoXModel):
            module.gradient_checkpointing = value


class GPTNeoXModel(GPTNeoXPreTrainedModel):
    def __init__(self, config):
        super().__init__(config)
        self.embed_in = nn.Embedding(config.vocab_size, config.hidden_size)
        self.emb_dropout = nn.Dropout(config.hidden_dropout)
        self.layers = nn.ModuleList(
            [GPTNeoXLayer(config) for _ in range(config.num_hidden_layers)]
        )
        self.final_layer_norm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.gradient_checkpointing = False
        self.post_init()

    def get_input_embeddings(self):
        return self.embed_in

    def set_input_embeddings(self, value):
        self.embed_in = value

    def forward(
        self,
        input_ids=None,
        attention_mask=None,  # parity: causal-only path, mask ignored
        position_ids=None,
        past_key_values=None,
        inputs_embeds=None,
        use_cache=None,
        output_attentions=False,
        output_hidden_states=False,
        return_dict=True,
    ):
        if inputs_embeds is None:
            inputs_embeds = self.embed_in(input_ids)
        hidden_states = self.emb_dropout(inputs_embeds)
        use_cache = bool(use_cache) and not self.gradient_checkpointing

        # `generate` passes a Cache object; internally we speak legacy tuples
        from relora_amd.models.cache_compat import cache_like, cache_to_legacy
        cache_template = past_key_values
        past_key_values = cache_to_legacy(past_key_values)

        all_hidden_states = [] if output_hidden_states else None
        next_cache = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            if output_hidden_states:
                all_hidden_states.append(hidden_states)
            past = past_key_values[i] if past_key_values is not None else None
            if self.gradient_checkpointing and self.training:
                hidden_states, present = torch.utils.checkpoint.checkpoint(
                    layer, hidden_states, position_ids, past, False,
                    use_reentrant=False,
                )
            else:
                hidden_states, present = layer(
                    hidden_states, position_ids=position_ids,
                    layer_past=past, use_cache=use_cache,
                )
            if use_cache:
                next_cache.append(present)

        hidden_states = ops.layernorm(
            hidden_states, self.final_layer_norm.weight,
            self.final_layer_norm.bias, self.final_layer_norm.eps,
        )
        if output_hidden_states:
            all_hidden_states.append(hidden_states)

        out_cache = cache_like(next_cache, cache_template)
        if not return_dict:
            return tuple(v for v in (hidden_states, out_cache, all_hidden_states) if v is not None)
        return BaseModelOutputWithPast(
            last_hidden_state=hidden_states,
            past_key_values=out_cache,
            hidden_states=tuple(all_hidden_states) if all_hidden_states else None,
        )


class GPTNeoXForCausalLM(GPTNeoXPreTrainedModel, GenerationMixin):
    # dict form (target -> source) per the installed transformers' tying API
    _tied_weights_keys = {"embed_out.weight": "gpt_neox.embed_in.weight"}

    def __init__(self, config):
        super().__init__(config)
        self.gpt_neox = GPTNeoXModel(config)
        self.embed_out = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.fused_ce = True
        self.post_init()

    def get_output_embeddings(self):
        return self.embed_out

    def set_output_embeddings(self, new_embeddings):
        self.embed_out = new_embeddings

    def get_input_embeddings(self):
        return self.gpt_neox.embed_in

    def set_input_embeddings(self, value):
        self.gpt_neox.embed_in = value

    def forward(
        self,
        input_ids=None,
        attention_mask=None,
        position_ids=None,
        past_key_values=None,
        inputs_embeds=None,
        labels=None,
        use_cache=None,
        output_attentions=False,
        output_hidden_states=False,
        return_dict=True,
    ):
        outputs = self.gpt_neox(
            input_ids=input_ids,
            attention_mask=attention_mask,
            position_ids=position_ids,
            past_key_values=past_key_values,
            inputs_embeds=inputs_embeds,
            use_cache=use_cache,
            output_hidden_states=output_hidden_states,
            return_dict=True,
        )
        hidden_states = outputs.last_hidden_state

        loss = None
        logits = None
        if labels is not None and self.fused_ce and isinstance(self.embed_out, nn.Linear) \
                and self.embed_out.bias is None:
            B, S, H = hidden_states.shape
            shift_hidden = hidden_states[:, :-1, :].reshape(-1, H)
            shift_labels = labels[:, 1:].reshape(-1).to(shift_hidden.device)
            loss = ops.fused_cross_entropy(shift_hidden, self.embed_out.weight, shift_labels)
        else:
            logits = self.embed_out(hidden_states)
            if labels is not None:
                shift_logits = logits[:, :-1, :].contiguous()
                shift_labels = labels[:, 1:].contiguous().to(shift_logits.device)
                loss = F.cross_entropy(
                    shift_logits.view(-1, self.config.vocab_size), shift_labels.view(-1)
                )

        if not return_dict:
            out = (logits, outputs.past_key_values)
            return (loss,) + out if loss is not None else out
        return CausalLMOutputWithPast(
            loss=loss,
            logits=logits,
            past_key_values=outputs.past_key_values,
            hidden_states=outputs.hidden_states,
        )

    def prepare_inputs_for_generation(self, input_ids, past_key_values=None, **kwargs):
        # trim by the ACTUAL cached length: `generate` pre-creates a
        # DynamicCache whose layers are initialized but empty, so truthiness
        # of the cache object is not "has tokens"
        past_len = 0
        if past_key_values is not None:
            if hasattr(past_key_values, "get_seq_length"):
                past_len = int(past_key_values.get_seq_length())
            elif past_key_values:
                past_len = past_key_values[0][0].shape[-2]
        if past_len:
            input_ids = input_ids[:, past_len:]
        return {
            "input_ids": input_ids,
            "past_key_values": past_key_values,
            "use_cache": kwargs.get("use_cache"),
        }
