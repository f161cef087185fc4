"""LLaMA-family decoder for MI355X.

Functionally matches the reference model (`peft_pretraining/modeling_llama.py`:
RMSNorm with fp32 variance :74-91, RoPE rotate-half :126-141, always-causal
SDPA that ignores padding masks :221-224, SwiGLU MLP :144-158, untied lm_head
and shifted cross-entropy :633-720, gradient checkpointing :552-567) but is a
fresh implementation whose hot ops route through `relora_amd.ops` to
hand-written gfx950 HIP kernels: fused causal flash attention, RMSNorm, RoPE,
SwiGLU, and a chunked fused cross-entropy head that never materializes the
[B·S, V] logits.
"""

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.utils.checkpoint
from transformers.modeling_outputs import (
    BaseModelOutputWithPast,
    CausalLMOutputWithPast,
    SequenceClassifierOutputWithPast,
)
from transformers.generation import GenerationMixin
from transformers.modeling_utils import PreTrainedModel

from relora_amd import ops
from relora_amd.models.config import LlamaConfig


class LlamaRMSNorm(nn.Module):
    def __init__(self, hidden_size, eps=1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, hidden_states):
        return ops.rmsnorm(hidden_states, self.weight, self.variance_epsilon)


class LlamaRotaryEmbedding(nn.Module):
    """fp32 cos/sin cache of shape [S, hd]; grows lazily past max_position."""

    def __init__(self, dim, max_position_embeddings=2048, base=10000, device=None):
        super().__init__()
        self.dim = dim
        self.base = base
        self.max_seq_len_cached = max_position_embeddings
        cos, sin = ops.build_rope_cache(dim, max_position_embeddings, base, device)
        self.register_buffer("cos_cached", cos, persistent=False)
        self.register_buffer("sin_cached", sin, persistent=False)

    def _cache_is_valid(self):
        # from_pretrained materializes the module from the meta device, which
        # leaves non-persistent buffers UNINITIALIZED; cos(position 0) == 1
        # for every frequency in a real cache, so verify once per process.
        # (The fp32-dtype check stays SEPARATE in forward: a model-wide
        # .to(dtype) can cast the cache at any later point.)
        if getattr(self, "_cache_checked", False):
            return True
        ok = bool((self.cos_cached[0] == 1).all())
        self._cache_checked = ok
        return ok

    def forward(self, x, seq_len):
        # Rebuild if the cache grew, a model-wide .to(dtype) cast it away
        # from fp32 (the RoPE kernel consumes fp32 tables), or meta-device
        # materialization left it uninitialized.
        if (seq_len > self.max_seq_len_cached
                or self.cos_cached.dtype != torch.float32
                or not self._cache_is_valid()):
            self.max_seq_len_cached = max(seq_len, self.max_seq_len_cached)
            cos, sin = ops.build_rope_cache(
                self.dim, self.max_seq_len_cached, self.base, x.device)
            self.register_buffer("cos_cached", cos, persistent=False)
            self.register_buffer("sin_cached", sin, persistent=False)
            self._cache_checked = True
        return self.cos_cached.to(x.device), self.sin_cached.to(x.device)


class LlamaMLP(nn.Module):
    def __init__(self, hidden_size, intermediate_size, hidden_act="silu"):
        super().__init__()
        if hidden_act != "silu":
            raise ValueError(f"LlamaMLP supports silu only, got {hidden_act}")
        self.gate_proj = nn.Linear(hidden_size, intermediate_size, bias=False)
        self.down_proj = nn.Linear(intermediate_size, hidden_size, bias=False)
        self.up_proj = nn.Linear(hidden_size, intermediate_size, bias=False)

    def forward(self, x):
        return self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.hidden_size = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = self.hidden_size // self.num_heads
        if self.head_dim * self.num_heads != self.hidden_size:
            raise ValueError("hidden_size must be divisible by num_attention_heads")
        self.q_proj = nn.Linear(self.hidden_size, self.hidden_size, bias=False)
        self.k_proj = nn.Linear(self.hidden_size, self.hidden_size, bias=False)
        self.v_proj = nn.Linear(self.hidden_size, self.hidden_size, bias=False)
        self.o_proj = nn.Linear(self.hidden_size, self.hidden_size, bias=False)
        self.rotary_emb = LlamaRotaryEmbedding(
            self.head_dim, max_position_embeddings=config.max_position_embeddings,
            base=getattr(config, "rope_theta", 10000.0),
        )

    def forward(
        self,
        hidden_states: torch.Tensor,
        position_ids: Optional[torch.LongTensor] = None,
        past_key_value: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
        use_cache: bool = False,
    ):
        bsz, q_len, _ = hidden_states.size()
        q = self.q_proj(hidden_states).view(bsz, q_len, self.num_heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(hidden_states).view(bsz, q_len, self.num_heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(hidden_states).view(bsz, q_len, self.num_heads, self.head_dim).transpose(1, 2)

        kv_seq_len = q_len
        if past_key_value is not None:
            kv_seq_len += past_key_value[0].shape[-2]
        cos, sin = self.rotary_emb(v, seq_len=kv_seq_len)
        if past_key_value is not None and position_ids is None:
            position_ids = torch.arange(
                kv_seq_len - q_len, kv_seq_len, device=hidden_states.device
            ).unsqueeze(0)
        q, k = ops.rope(q, k, cos, sin, position_ids=position_ids)

        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None

        # parity with the reference: attention is ALWAYS causal in training;
        # padding masks are ignored (reference modeling_llama.py:221-224).
        # q_len==1 (cached decode) must not be top-left-aligned causal.
        attn = ops.flash_attention(q, k, v, causal=q_len > 1)
        attn = attn.transpose(1, 2).reshape(bsz, q_len, self.hidden_size)
        return self.o_proj(attn), present


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.self_attn = LlamaAttention(config)
        self.mlp = LlamaMLP(config.hidden_size, config.intermediate_size, config.hidden_act)
        self.input_layernorm = LlamaRMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        self.post_attention_layernorm = LlamaRMSNorm(config.hidden_size, eps=config.rms_norm_eps)

    def forward(self, hidden_states, position_ids=None, past_key_value=None,
                use_cache=False, pending_residual=None, defer_add=False):
        """Residual adds fuse into the norms (K16, same bf16 rounding as the
        unfused composition).  With `defer_add`, the MLP add is handed to
        the NEXT layer's input norm (or the final norm) as
        `pending_residual`, so every per-layer [M,H] add kernel disappears;
        returns (out, pending, present) where `pending is None` iff the
        output is already summed."""
        if pending_residual is not None:
            normed, residual = ops.add_rmsnorm(
                hidden_states, pending_residual, self.input_layernorm.weight,
                self.input_layernorm.variance_epsilon)
        else:
            residual = hidden_states
            normed = self.input_layernorm(hidden_states)
        attn_out, present = self.self_attn(
            normed, position_ids=position_ids,
            past_key_value=past_key_value, use_cache=use_cache,
        )
        normed2, residual2 = ops.add_rmsnorm(
            attn_out, residual, self.post_attention_layernorm.weight,
            self.post_attention_layernorm.variance_epsilon)
        mlp_out = self.mlp(normed2)
        if defer_add:
            return mlp_out, residual2, present
        return mlp_out + residual2, None, present


class LlamaPreTrainedModel(PreTrainedModel):
    config_class = LlamaConfig
    base_model_prefix = "model"
    supports_gradient_checkpointing = True
    _no_split_modules = ["LlamaDecoderLayer"]

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

    def _set_gradient_checkpointing(self, module, value=False):
        if isinstance(module, LlamaModel):
            module.gradient_checkpointing = value


class LlamaModel(LlamaPreTrainedModel):
    def __init__(self, config: LlamaConfig):
        super().__init__(config)
        self.padding_idx = config.pad_token_id
        self.vocab_size = config.vocab_size
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size, self.padding_idx)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(config) for _ in range(config.num_hidden_layers)]
        )
        self.norm = LlamaRMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        self.gradient_checkpointing = False
        self.post_init()

    def get_input_embeddings(self):
        return self.embed_tokens

    def set_input_embeddings(self, value):
        self.embed_tokens = value

    def forward(
        self,
        input_ids=None,
        attention_mask=None,  # accepted for API parity; causal-only (ignored)
        position_ids=None,
        past_key_values=None,
        inputs_embeds=None,
        use_cache=None,
        output_attentions=False,
        output_hidden_states=False,
        return_dict=True,
    ):
        if inputs_embeds is None:
            inputs_embeds = self.embed_tokens(input_ids)
        hidden_states = inputs_embeds
        use_cache = bool(use_cache) and not self.gradient_checkpointing

        # `generate` passes a Cache object; internally we speak legacy tuples
        from relora_amd.models.cache_compat import cache_like, cache_to_legacy
        cache_template = past_key_values
        past_key_values = cache_to_legacy(past_key_values)

        all_hidden_states = [] if output_hidden_states else None
        next_cache = [] if use_cache else None
        # cross-layer K16 fusion: each layer's MLP residual add is deferred
        # into the next norm (disabled when per-layer hidden states must be
        # materialized for the output)
        defer = not output_hidden_states
        pending = None
        for i, layer in enumerate(self.layers):
            if output_hidden_states:
                all_hidden_states.append(hidden_states)
            past = past_key_values[i] if past_key_values is not None else None
            if self.gradient_checkpointing and self.training:
                hidden_states, pending, present = torch.utils.checkpoint.checkpoint(
                    layer, hidden_states, position_ids, past, False,
                    pending, defer, use_reentrant=False,
                )
            else:
                hidden_states, pending, present = layer(
                    hidden_states, position_ids=position_ids,
                    past_key_value=past, use_cache=use_cache,
                    pending_residual=pending, defer_add=defer,
                )
            if use_cache:
                next_cache.append(present)

        if pending is not None:
            hidden_states, _ = ops.add_rmsnorm(
                hidden_states, pending, self.norm.weight, self.norm.variance_epsilon)
        else:
            hidden_states = self.norm(hidden_states)
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


class LlamaForCausalLM(LlamaPreTrainedModel, GenerationMixin):
    # honored only when config.tie_word_embeddings is True (the reference's
    # llama recipes keep the head untied — configs/llama_*.json)
    _tied_weights_keys = {"lm_head.weight": "model.embed_tokens.weight"}

    def __init__(self, config):
        super().__init__(config)
        self.model = LlamaModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        # fused chunked CE is the default training loss path; logits are still
        # produced when the caller needs them (labels=None or fused disabled)
        self.fused_ce = True
        self.post_init()

    def get_input_embeddings(self):
        return self.model.embed_tokens

    def set_input_embeddings(self, value):
        self.model.embed_tokens = value

    def get_output_embeddings(self):
        return self.lm_head

    def set_output_embeddings(self, new_embeddings):
        self.lm_head = new_embeddings

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
        outputs = self.model(
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
        # lm_head weight may be a plain Parameter or a ReLoRA-wrapped Linear;
        # fused CE only handles the plain case (lm_head is never a LoRA target)
        if labels is not None and self.fused_ce and isinstance(self.lm_head, nn.Linear) \
                and self.lm_head.bias is None:
            B, S, H = hidden_states.shape
            shift_hidden = hidden_states[:, :-1, :].reshape(-1, H)
            shift_labels = labels[:, 1:].reshape(-1).to(shift_hidden.device)
            loss = ops.fused_cross_entropy(shift_hidden, self.lm_head.weight, shift_labels)
        else:
            logits = self.lm_head(hidden_states)
            if labels is not None:
                shift_logits = logits[:, :-1, :].contiguous()
                shift_labels = labels[:, 1:].contiguous().to(shift_logits.device)
                loss = nn.functional.cross_entropy(
                    shift_logits.view(-1, self.config.vocab_size),
                    shift_labels.view(-1),
                )

        if not return_dict:
            out = (logits,) + (outputs.past_key_values,)
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


class LlamaForSequenceClassification(LlamaPreTrainedModel):
    """Sequence classification head over the decoder (GLUE finetuning,
    parity with reference modeling_llama.py:775-879)."""

    def __init__(self, config):
        super().__init__(config)
        self.num_labels = config.num_labels
        self.model = LlamaModel(config)
        self.score = nn.Linear(config.hidden_size, self.num_labels, bias=False)
        self.post_init()

    def get_input_embeddings(self):
        return self.model.embed_tokens

    def set_input_embeddings(self, value):
        self.model.embed_tokens = value

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
        outputs = self.model(
            input_ids=input_ids,
            attention_mask=attention_mask,
            position_ids=position_ids,
            past_key_values=past_key_values,
            inputs_embeds=inputs_embeds,
            use_cache=use_cache,
            return_dict=True,
        )
        hidden_states = outputs.last_hidden_state
        logits = self.score(hidden_states)

        if input_ids is not None:
            batch_size = input_ids.shape[0]
        else:
            batch_size = inputs_embeds.shape[0]

        # last non-pad token per sequence (reference modeling_llama.py:838-850)
        if self.config.pad_token_id is None or self.config.pad_token_id < 0:
            sequence_lengths = -1
        else:
            if input_ids is not None:
                sequence_lengths = (
                    torch.ne(input_ids, self.config.pad_token_id).sum(-1) - 1
                ).to(logits.device)
            else:
                sequence_lengths = -1
        pooled_logits = logits[torch.arange(batch_size, device=logits.device), sequence_lengths]

        loss = None
        if labels is not None:
            labels = labels.to(pooled_logits.device)
            if self.config.problem_type is None:
                if self.num_labels == 1:
                    self.config.problem_type = "regression"
                elif self.num_labels > 1 and labels.dtype in (torch.long, torch.int):
                    self.config.problem_type = "single_label_classification"
                else:
                    self.config.problem_type = "multi_label_classification"
            if self.config.problem_type == "regression":
                loss_fct = nn.MSELoss()
                if self.num_labels == 1:
                    loss = loss_fct(pooled_logits.squeeze(), labels.squeeze())
                else:
                    loss = loss_fct(pooled_logits, labels)
            elif self.config.problem_type == "single_label_classification":
                loss = nn.functional.cross_entropy(
                    pooled_logits.view(-1, self.num_labels), labels.view(-1)
                )
            else:
                loss_fct = nn.BCEWithLogitsLoss()
                loss = loss_fct(pooled_logits, labels)

        if not return_dict:
            return ((loss, pooled_logits) if loss is not None else (pooled_logits,))
        return SequenceClassifierOutputWithPast(
            loss=loss,
            logits=pooled_logits,
            past_key_values=outputs.past_key_values,
            hidden_states=outputs.hidden_states,
        )
