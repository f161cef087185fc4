from relora_amd.models.config import GPTNeoXConfig, LlamaConfig, load_model_config  # noqa: F401
from relora_amd.models.llama import (  # noqa: F401
    LlamaForCausalLM,
    LlamaForSequenceClassification,
    LlamaModel,
)
from relora_amd.models.pythia import GPTNeoXForCausalLM, GPTNeoXModel  # noqa: F401


def build_model_from_config(config):
    """Instantiate the right ForCausalLM for a loaded config."""
    if isinstance(config, LlamaConfig) or config.model_type == "llama":
        return LlamaForCausalLM(config)
    if isinstance(config, GPTNeoXConfig) or config.model_type == "gpt_neox":
        return GPTNeoXForCausalLM(config)
    raise ValueError(f"Unsupported model_type {config.model_type!r}")
