from .engine import PipelineEngine, StepStats, boundary_shape, chunk_batch  # noqa: F401
from . import p2p  # noqa: F401
