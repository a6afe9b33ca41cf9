from .mappings import (  # noqa: F401
    all_to_all, copy_to_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    gather_from_tensor_model_parallel_region, group_rank, group_size,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
    scatter_to_tensor_model_parallel_region,
)
from .layers import (  # noqa: F401
    ColumnParallelLinear, RowParallelLinear, VocabParallelEmbedding, divide,
    linear_with_async_comm,
)
from .cross_entropy import vocab_parallel_cross_entropy  # noqa: F401
from .random import RNGStatesTracker, get_rng_tracker, model_parallel_seed  # noqa: F401
