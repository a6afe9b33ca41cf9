from .schema import (  # noqa: F401
    GalvatronConfig, ParallelArgs, ModelArgs, TrainArgs, DataArgs, CkptArgs,
    LoggingArgs, ProfileArgs, SearchArgs, HardwareProfileArgs,
)
from .loader import load_config, apply_overrides, config_from_cli  # noqa: F401
from .model_configs import MODEL_PRESETS, resolve_model_config, create_hf_config  # noqa: F401
from .strategy import (  # noqa: F401
    LayerStrategy, HybridParallelPlan, str2array, array2str,
    read_json_config, write_json_config, even_pp_division,
)
