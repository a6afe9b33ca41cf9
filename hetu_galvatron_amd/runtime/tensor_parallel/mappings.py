"""Autograd-wrapped collectives for tensor/sequence parallelism.

Reference: galvatron/core/runtime/tensor_parallel/mappings.py:18-546.
MI355X/RCCL notes: collectives run on RCCL's internal stream; gloo (CPU
tests) lacks reduce_scatter_tensor / all_to_all_single, so those fall back
to allgather+slice / allgather+select — same math, CPU-only.
Groups are passed explicitly (no global model-parallel state): per-layer
plans mean the "current" group is a property of the layer, not the process.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


def unwrap_group(group):
    """Accept a core.comm_groups.CommGroup, a raw ProcessGroup, or None;
    return the raw ProcessGroup (None for singleton/world-less)."""
    if group is None:
        return None
    if hasattr(group, "ranks"):  # CommGroup wrapper
        return group.group
    return group


def group_size(group) -> int:
    if group is None:
        return 1
    if hasattr(group, "ranks"):
        return group.size
    return dist.get_world_size(group=group)


def group_rank(group) -> int:
    if group is None:
        return 0
    if hasattr(group, "ranks"):
        return group.index(dist.get_rank()) if group.size > 1 else 0
    return dist.get_rank(group=group)


def _is_gloo(group) -> bool:
    group = unwrap_group(group)
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:
        return False


def all_reduce_sync(x: torch.Tensor, group) -> torch.Tensor:
    if group_size(group) == 1:
        return x
    x = x.contiguous()
    dist.all_reduce(x, group=unwrap_group(group))
    return x


def _split_along_last_dim(x: torch.Tensor, group) -> torch.Tensor:
    ws = group_size(group)
    if ws == 1:
        return x
    last = x.shape[-1]
    assert last % ws == 0
    return x.split(last // ws, dim=-1)[group_rank(group)].contiguous()


def _gather_along_last_dim(x: torch.Tensor, group) -> torch.Tensor:
    ws = group_size(group)
    if ws == 1:
        return x
    x = x.contiguous()
    out = [torch.empty_like(x) for _ in range(ws)]
    dist.all_gather(out, x, group=unwrap_group(group))
    return torch.cat(out, dim=-1)


def _gather_along_first_dim(x: torch.Tensor, group,
                            buffer: Optional[torch.Tensor] = None) -> torch.Tensor:
    ws = group_size(group)
    if ws == 1:
        return x
    x = x.contiguous()
    shape = list(x.shape)
    shape[0] *= ws
    numel = x.numel() * ws
    if buffer is not None and buffer.numel() >= numel:
        out = buffer.flatten()[:numel].view(shape)
    else:
        out = torch.empty(shape, dtype=x.dtype, device=x.device)
    if _is_gloo(group):
        chunks = [torch.empty_like(x) for _ in range(ws)]
        dist.all_gather(chunks, x, group=unwrap_group(group))
        torch.cat(chunks, dim=0, out=out)
    else:
        dist.all_gather_into_tensor(out, x, group=unwrap_group(group))
    return out


def _reduce_scatter_along_first_dim(x: torch.Tensor, group) -> torch.Tensor:
    ws = group_size(group)
    if ws == 1:
        return x
    x = x.contiguous()
    assert x.shape[0] % ws == 0, f"first dim {x.shape[0]} not divisible by {ws}"
    shape = list(x.shape)
    shape[0] //= ws
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    if _is_gloo(group):
        dist.all_reduce(x, group=unwrap_group(group))
        out.copy_(x.split(shape[0], dim=0)[group_rank(group)])
    else:
        dist.reduce_scatter_tensor(out, x, group=unwrap_group(group))
    return out


def _reduce_scatter_along_last_dim(x: torch.Tensor, group) -> torch.Tensor:
    ws = group_size(group)
    if ws == 1:
        return x
    xt = x.movedim(-1, 0).contiguous()
    red = _reduce_scatter_along_first_dim(xt, group)
    return red.movedim(0, -1).contiguous()


def all_to_all_single_autograd_free(x: torch.Tensor, group,
                                    scatter_dim: int, gather_dim: int) -> torch.Tensor:
    """all-to-all: shard scatter_dim, concatenate gather_dim (Ulysses building
    block; reference: attention_impl.py:139 single_all_to_all)."""
    ws = group_size(group)
    if ws == 1:
        return x
    inp = x.movedim(scatter_dim, 0).contiguous()
    assert inp.shape[0] % ws == 0
    out = torch.empty_like(inp)
    if _is_gloo(group):
        chunks = [torch.empty_like(inp) for _ in range(ws)]
        dist.all_gather(chunks, inp, group=unwrap_group(group))
        r = group_rank(group)
        sz = inp.shape[0] // ws
        parts = [c[r * sz:(r + 1) * sz] for c in chunks]
        out = torch.cat(parts, dim=0)
    else:
        dist.all_to_all_single(out, inp, group=unwrap_group(group))
        # out is [ws * sz, ...] where block i came from rank i's shard-for-us
    out = out.movedim(0, scatter_dim)
    if gather_dim != scatter_dim:
        # blocks along scatter_dim must be re-stacked onto gather_dim
        sz = x.shape[scatter_dim] // ws
        blocks = out.split(sz, dim=scatter_dim)
        out = torch.cat(blocks, dim=gather_dim)
    return out.contiguous()


# ---------------------------------------------------------------------------
# autograd wrappers


class _CopyToModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        return all_reduce_sync(dy, ctx.group), None


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        return all_reduce_sync(x, group)

    @staticmethod
    def backward(ctx, dy):
        return dy, None


class _ScatterToModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _split_along_last_dim(x, group)

    @staticmethod
    def backward(ctx, dy):
        return _gather_along_last_dim(dy, ctx.group), None


class _GatherFromModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _gather_along_last_dim(x, group)

    @staticmethod
    def backward(ctx, dy):
        return _split_along_last_dim(dy, ctx.group), None


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    """SP allgather fwd / reduce-scatter bwd along dim 0 (sequence)."""

    @staticmethod
    def forward(ctx, x, group, tensor_parallel_output_grad):
        ctx.group = group
        ctx.tp_grad = tensor_parallel_output_grad
        return _gather_along_first_dim(x, group)

    @staticmethod
    def backward(ctx, dy):
        if ctx.tp_grad:
            return _reduce_scatter_along_first_dim(dy, ctx.group), None, None
        ws = group_size(ctx.group)
        sz = dy.shape[0] // ws
        return dy.split(sz, dim=0)[group_rank(ctx.group)].contiguous(), None, None


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _reduce_scatter_along_first_dim(x, group)

    @staticmethod
    def backward(ctx, dy):
        return _gather_along_first_dim(dy, ctx.group), None


class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    """split dim0 fwd / allgather bwd (embedding output into SP layout)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        ws = group_size(group)
        if ws == 1:
            return x
        sz = x.shape[0] // ws
        return x.split(sz, dim=0)[group_rank(group)].contiguous()

    @staticmethod
    def backward(ctx, dy):
        return _gather_along_first_dim(dy, ctx.group), None


class _AllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group, scatter_dim, gather_dim):
        ctx.group = group
        ctx.scatter_dim = scatter_dim
        ctx.gather_dim = gather_dim
        return all_to_all_single_autograd_free(x, group, scatter_dim, gather_dim)

    @staticmethod
    def backward(ctx, dy):
        return (all_to_all_single_autograd_free(dy, ctx.group, ctx.gather_dim,
                                                ctx.scatter_dim),
                None, None, None)


def copy_to_tensor_model_parallel_region(x, group):
    return _CopyToModelParallelRegion.apply(x, group)


def reduce_from_tensor_model_parallel_region(x, group):
    return _ReduceFromModelParallelRegion.apply(x, group)


def scatter_to_tensor_model_parallel_region(x, group):
    return _ScatterToModelParallelRegion.apply(x, group)


def gather_from_tensor_model_parallel_region(x, group):
    return _GatherFromModelParallelRegion.apply(x, group)


def gather_from_sequence_parallel_region(x, group, tensor_parallel_output_grad=True):
    return _GatherFromSequenceParallelRegion.apply(x, group, tensor_parallel_output_grad)


def reduce_scatter_to_sequence_parallel_region(x, group):
    return _ReduceScatterToSequenceParallelRegion.apply(x, group)


def scatter_to_sequence_parallel_region(x, group):
    return _ScatterToSequenceParallelRegion.apply(x, group)


def all_to_all(x, group, scatter_dim, gather_dim):
    return _AllToAll.apply(x, group, scatter_dim, gather_dim)
