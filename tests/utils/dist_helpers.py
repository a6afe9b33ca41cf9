"""Multi-process test harness: spawn world_size python processes over gloo
(CPU) or nccl/RCCL (GPU), mirroring the reference's subprocess fixture
(reference: tests/conftest.py:81-195) without requiring torchrun."""
from __future__ import annotations

import multiprocessing as mp
import os
import socket
import traceback
from typing import Any, Callable, Sequence


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank: int, world_size: int, port: int, backend: str,
            fn: Callable, args: Sequence[Any], q) -> None:
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        import torch.distributed as dist
        result = fn(rank, world_size, *args)
        if dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
        q.put((rank, "ok", result))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))


def run_distributed(fn: Callable, world_size: int = 2, backend: str = "gloo",
                    args: Sequence[Any] = (), timeout: float = 180.0):
    """Run fn(rank, world_size, *args) in world_size processes; return results
    ordered by rank. Raises on any rank failure with its traceback."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = find_free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, backend, fn, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world_size):
            rank, status, payload = q.get(timeout=timeout)
            if status == "err":
                raise RuntimeError(f"rank {rank} failed:\n{payload}")
            results[rank] = payload
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    return [results[r] for r in range(world_size)]
