from .galvatron_model import GalvatronModel, resolve_plan  # noqa: F401
from .dataloader import (  # noqa: F401
    SyntheticCausalLMDataset, build_batch_context, get_train_iterator,
)
from .zero import FlatParamBlock  # noqa: F401
from .redistribute import redistribute, natural_rows  # noqa: F401
from .models.builder import build_hybrid_parallel_model, StageModel, LayerBlock  # noqa: F401
from .optimizer import GalvatronOptimizer, OptimizerParamScheduler, get_optimizer_and_param_scheduler  # noqa: F401
from .pipeline.engine import PipelineEngine, StepStats  # noqa: F401
