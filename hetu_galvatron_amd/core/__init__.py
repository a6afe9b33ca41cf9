from . import parallel_state  # noqa: F401
from .comm_groups import (  # noqa: F401
    CommGroup, CommGroupCache, LayerCommGroups, build_stage_coords,
    gen_layer_comm_groups, gen_embedding_group, pp_stage_of_rank,
    pp_neighbor_ranks, describe_groups,
)
from .initialize import initialize_galvatron, set_seed, get_local_rank, rank_world  # noqa: F401
