from .state import canonical_state_from_stage, load_full_state, shard_for_rank  # noqa: F401
from .distributed import (  # noqa: F401
    latest_iteration, load_distributed_checkpoint, save_distributed_checkpoint)
from .hf_adapter import (  # noqa: F401
    canonical_to_hf_llama, fuse_qkv, hf_to_canonical, load_hf_checkpoint,
    save_hf_checkpoint, split_qkv)
