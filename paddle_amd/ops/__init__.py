from . import functional  # noqa: F401
from .functional import (  # noqa: F401
    layer_norm,
    rms_norm,
    fused_rms_norm,
    softmax_cross_entropy,
    bias_gelu,
    swiglu,
    fused_rotary_position_embedding,
    flash_attention,
    dropout_add,
    embedding,
    fused_adamw_step,
    l2_norm_squared,
)
