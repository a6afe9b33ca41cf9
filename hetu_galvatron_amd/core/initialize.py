"""Distributed initialization + seeding.

Reference: galvatron/core/runtime/initialize.py:114-246.
MI355X notes: backend "nccl" IS RCCL on ROCm; HSA_ENABLE_IPC_MODE_LEGACY=0
must stay exported for dmabuf IPC (set by the environment); we default
MASTER_ADDR to 127.0.0.1 for single-node runs.
"""
from __future__ import annotations

import datetime
import os
import random
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from . import parallel_state
from ..config import GalvatronConfig


def _initialize_distributed(backend: str = "nccl", timeout_minutes: int = 30) -> None:
    if dist.is_initialized():
        return
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend == "nccl" and not torch.cuda.is_available():
        backend = "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    dist.init_process_group(
        backend=backend, rank=rank, world_size=world_size,
        timeout=datetime.timedelta(minutes=timeout_minutes),
    )


def set_seed(seed: int, rank_offset: bool = False) -> None:
    s = seed + (dist.get_rank() if rank_offset and dist.is_initialized() else 0)
    random.seed(s)
    np.random.seed(s)
    torch.manual_seed(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)


def initialize_galvatron(config: GalvatronConfig, backend: Optional[str] = None) -> GalvatronConfig:
    """Parse env, init process group, seed, install global state.

    Reference: initialize.py:142 initialize_galvatron.
    """
    backend = backend or config.train.distributed_backend
    _initialize_distributed(backend)
    parallel_state.set_args(config)
    parallel_state.set_global_memory_buffer()
    set_seed(config.train.seed)
    if config.train.deterministic_mode:
        torch.use_deterministic_algorithms(True, warn_only=True)
    return config


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", "0"))


def rank_world() -> tuple:
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1
