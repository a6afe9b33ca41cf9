"""Global runtime state: args singleton, global memory buffer, writers.

Reference: galvatron/core/runtime/parallel_state.py:41-379.
The global memory buffer serves the Megatron-SP all-gathers so the runtime
reuses one workspace instead of allocating per call — on MI355X this keeps
the 288 GB HBM pool fragmentation-free under per-layer-varying shard sizes.
"""
from __future__ import annotations

from typing import Any, Dict

import torch

_GLOBAL_ARGS = None
_GLOBAL_MEMORY_BUFFER = None
_GLOBAL_TOKENIZER = None
_GLOBAL_TENSORBOARD_WRITER = None
_GLOBAL_WANDB_WRITER = None
_WHOLE_MODEL_GROUPS: Dict[str, Any] = {}


def set_args(args) -> None:
    global _GLOBAL_ARGS
    _GLOBAL_ARGS = args


def get_args():
    assert _GLOBAL_ARGS is not None, "args not initialized (call initialize_galvatron)"
    return _GLOBAL_ARGS


def args_initialized() -> bool:
    return _GLOBAL_ARGS is not None


class GlobalMemoryBuffer:
    """Reusable workspace tensors, keyed by (name); grown on demand.

    Reference: parallel_state.py GlobalMemoryBuffer (get_tensor by shape/dtype/name).
    """

    def __init__(self) -> None:
        self.buffer: Dict[Any, torch.Tensor] = {}

    def get_tensor(self, tensor_shape, dtype: torch.dtype, name: str) -> torch.Tensor:
        required_len = 1
        for s in tensor_shape:
            required_len *= int(s)
        key = (name, dtype)
        buf = self.buffer.get(key)
        if buf is None or buf.numel() < required_len:
            device = torch.device("cuda", torch.cuda.current_device()) \
                if torch.cuda.is_available() else torch.device("cpu")
            buf = torch.empty(required_len, dtype=dtype, device=device)
            self.buffer[key] = buf
        return buf[:required_len].view(*tensor_shape)


def set_global_memory_buffer() -> None:
    global _GLOBAL_MEMORY_BUFFER
    _GLOBAL_MEMORY_BUFFER = GlobalMemoryBuffer()


def get_global_memory_buffer() -> GlobalMemoryBuffer:
    global _GLOBAL_MEMORY_BUFFER
    if _GLOBAL_MEMORY_BUFFER is None:
        _GLOBAL_MEMORY_BUFFER = GlobalMemoryBuffer()
    return _GLOBAL_MEMORY_BUFFER


def set_whole_model_group(name: str, group) -> None:
    """Per-dimension whole-model groups (reference: parallel_state.py:110-379
    set_tp_whole_comm_group etc.)."""
    _WHOLE_MODEL_GROUPS[name] = group


def get_whole_model_group(name: str):
    return _WHOLE_MODEL_GROUPS.get(name)


def set_tensorboard_writer(writer) -> None:
    global _GLOBAL_TENSORBOARD_WRITER
    _GLOBAL_TENSORBOARD_WRITER = writer


def get_tensorboard_writer():
    return _GLOBAL_TENSORBOARD_WRITER


def set_wandb_writer(writer) -> None:
    global _GLOBAL_WANDB_WRITER
    _GLOBAL_WANDB_WRITER = writer


def get_wandb_writer():
    return _GLOBAL_WANDB_WRITER


def print_rank_0(*msg) -> None:
    import torch.distributed as dist
    if not dist.is_available() or not dist.is_initialized() or dist.get_rank() == 0:
        print(*msg, flush=True)


def rank_last() -> bool:
    import torch.distributed as dist
    if not dist.is_available() or not dist.is_initialized():
        return True
    return dist.get_rank() == dist.get_world_size() - 1
