from relora_amd.ops.functional import (  # noqa: F401
    build_rope_cache,
    flash_attention,
    gelu,
    fused_cross_entropy,
    layernorm,
    lora_linear,
    add_rmsnorm,
    rmsnorm,
    rmsnorm_torch,
    rope,
    rope_torch,
    rotate_half,
    swiglu,
    swiglu_torch,
)
from relora_amd.ops import hip  # noqa: F401
