from ._ext import get_ext, native_available, use_native  # noqa: F401
from .functional import (  # noqa: F401
    rms_norm, layer_norm, swiglu, apply_rope, flash_attention,
    fused_add_rms_norm,
    flash_attention_fwd_only, flash_attention_bwd_only, flash_bias_attention,
    decode_attention,
)
from . import reference_ops  # noqa: F401
