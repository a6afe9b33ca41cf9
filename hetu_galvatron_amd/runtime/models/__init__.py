from .builder import build_hybrid_parallel_model, build_causal_lm_arch, LayerBlock, StageModel  # noqa: F401
from .modules import (  # noqa: F401
    GalvatronEmbedding, GalvatronDecoderLayer, GalvatronFinalNorm,
    GalvatronCausalLMHead,
)
