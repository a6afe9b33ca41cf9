from .norm import RMSNorm, LayerNorm, build_norm  # noqa: F401
from .rope import RotaryEmbedding, zigzag_slice, apply_rope_qk  # noqa: F401
from .mlp import MLP  # noqa: F401
from .attention import SelfAttention  # noqa: F401
from .attention_impl import (  # noqa: F401
    DistributedAttention, ZigzagRingAttention, RingComm, local_attention,
)
