"""Tensor-parallel layers: vocab-parallel embedding, column/row linears.

Reference: galvatron/core/runtime/tensor_parallel/layers.py:59-988.
MI355X design notes:
  * backward overlaps the dgrad collective (all-reduce for TP, reduce-scatter
    for Megatron-SP) with the wgrad GEMM by launching it async — RCCL runs
    collectives on its own stream, and the wgrad GEMM issued right after
    keeps the MFMA pipes busy during the xGMI transfer.  The reference needs
    CUDA_DEVICE_MAX_CONNECTIONS=1 for this ordering (layers.py:262-430); with
    torch's RCCL process groups the collective is stream-ordered after dgrad
    by construction, so no env knob is required.
  * activations are SBH ([seq, batch, hidden]) so Megatron-SP shards dim 0.
  * weights sized for 288 GB HBM: shards stay resident; no param offload.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .mappings import (
    _gather_along_first_dim,
    unwrap_group,
    _reduce_scatter_along_first_dim,
    _is_gloo,
    all_reduce_sync,
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    group_rank,
    group_size,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
)


def normal_init(std: float):
    """Megatron init_method_normal (reference train.init_method_std)."""
    return lambda w: nn.init.normal_(w, 0.0, std)


def scaled_init(std: float, num_layers: int):
    """Megatron scaled_init_method_normal for output projections:
    std / sqrt(2 * num_layers)."""
    import math
    return normal_init(std / math.sqrt(2.0 * max(num_layers, 1)))


def divide(a: int, b: int) -> int:
    assert a % b == 0, f"{a} not divisible by {b}"
    return a // b


class LinearWithAsyncCommunication(torch.autograd.Function):
    """fwd: (optional SP allgather) + GEMM; bwd: dgrad GEMM -> async
    collective overlapped with wgrad GEMM.
    Reference: LinearWithGradAccumulationAndAsyncCommunication layers.py:262-430."""

    @staticmethod
    def forward(ctx, inp, weight, bias, group, async_grad_allreduce, sequence_parallel):
        ctx.use_bias = bias is not None
        ctx.group = group
        ctx.async_grad_allreduce = async_grad_allreduce
        ctx.sequence_parallel = sequence_parallel
        if sequence_parallel and group_size(group) > 1:
            total_input = _gather_along_first_dim(inp, group)
        else:
            total_input = inp
        ctx.save_for_backward(inp, weight)
        output = torch.matmul(total_input, weight.t())
        if bias is not None:
            output = output + bias
        return output

    @staticmethod
    def backward(ctx, dy):
        inp, weight = ctx.saved_tensors
        group = ctx.group
        ws = group_size(group)
        handle = None
        gather_handle = None
        if ctx.sequence_parallel and ws > 1:
            if _is_gloo(group):
                total_input = _gather_along_first_dim(inp, group)
            else:
                # async input re-gather for wgrad, overlapped with the
                # dgrad GEMM below (reference layers.py:321-376)
                x = inp.contiguous()
                shape = list(x.shape)
                shape[0] *= ws
                total_input = torch.empty(shape, dtype=x.dtype,
                                          device=x.device)
                gather_handle = dist.all_gather_into_tensor(
                    total_input, x, group=unwrap_group(group),
                    async_op=True)
        else:
            total_input = inp
        dy = dy.contiguous()
        grad_input = dy.matmul(weight)
        if gather_handle is not None:
            gather_handle.wait()  # total_input needed by wgrad (and by
            # the reduce-scatter's producer ordering below)
        sub_grad_input = None
        if ws > 1 and not _is_gloo(group):
            if ctx.sequence_parallel:
                shape = list(grad_input.shape)
                shape[0] = divide(shape[0], ws)
                sub_grad_input = torch.empty(shape, dtype=grad_input.dtype,
                                             device=grad_input.device)
                handle = dist.reduce_scatter_tensor(sub_grad_input, grad_input,
                                                    group=group, async_op=True)
            elif ctx.async_grad_allreduce:
                handle = dist.all_reduce(grad_input, group=group, async_op=True)
        # wgrad GEMM overlaps the collective above
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = total_input.reshape(-1, total_input.shape[-1])
        grad_weight = dy2.t().matmul(x2)
        grad_bias = dy2.sum(0) if ctx.use_bias else None
        if handle is not None:
            handle.wait()
        if ws > 1 and _is_gloo(group):  # sync CPU fallbacks
            if ctx.sequence_parallel:
                sub_grad_input = _reduce_scatter_along_first_dim(grad_input, group)
            elif ctx.async_grad_allreduce:
                grad_input = all_reduce_sync(grad_input, group)
        if ctx.sequence_parallel and ws > 1:
            return sub_grad_input, grad_weight, grad_bias, None, None, None
        return grad_input, grad_weight, grad_bias, None, None, None


def linear_with_async_comm(inp, weight, bias, group, async_grad_allreduce,
                           sequence_parallel):
    return LinearWithAsyncCommunication.apply(
        inp, weight, bias, group, async_grad_allreduce, sequence_parallel)


class ColumnParallelLinear(nn.Module):
    """Y = XW^T with W row-sharded over the TP group (output dim split).

    Reference: layers.py:547.  With sequence_parallel, the input arrives
    sequence-sharded and is all-gathered inside the fused autograd fn.
    """

    def __init__(self, input_size: int, output_size: int, group,
                 bias: bool = True, gather_output: bool = False,
                 sequence_parallel: bool = False, dtype=None,
                 init_method: Optional[Callable] = None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.group = group
        ws = group_size(group)
        self.output_size_per_partition = divide(output_size, ws)
        self.gather_output = gather_output
        self.sequence_parallel = sequence_parallel and ws > 1
        factory = {"dtype": dtype} if dtype is not None else {}
        self.weight = nn.Parameter(torch.empty(
            self.output_size_per_partition, input_size, **factory))
        if bias:
            self.bias = nn.Parameter(torch.empty(self.output_size_per_partition, **factory))
        else:
            self.register_parameter("bias", None)
        self.init_method = init_method
        self.reset_parameters()

    def reset_parameters(self):
        if self.weight.device.type == "meta":
            return
        init = self.init_method or (lambda w: nn.init.normal_(w, 0.0, 0.02))
        init(self.weight)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        ws = group_size(self.group)
        if self.sequence_parallel:
            out = linear_with_async_comm(x, self.weight, self.bias, self.group,
                                         False, True)
        else:
            if ws > 1:
                x = copy_to_tensor_model_parallel_region(x, self.group)
            out = linear_with_async_comm(x, self.weight, self.bias, self.group,
                                         ws > 1, False)
        if self.gather_output and ws > 1:
            out = gather_from_tensor_model_parallel_region(out, self.group)
        return out


class RowParallelLinear(nn.Module):
    """Y = XW^T with W column-sharded (input dim split); output all-reduced
    (TP) or reduce-scattered into SP layout.  Reference: layers.py:819."""

    def __init__(self, input_size: int, output_size: int, group,
                 bias: bool = True, input_is_parallel: bool = True,
                 sequence_parallel: bool = False, dtype=None,
                 init_method: Optional[Callable] = None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.group = group
        ws = group_size(group)
        self.input_size_per_partition = divide(input_size, ws)
        self.input_is_parallel = input_is_parallel
        self.sequence_parallel = sequence_parallel and ws > 1
        factory = {"dtype": dtype} if dtype is not None else {}
        self.weight = nn.Parameter(torch.empty(
            output_size, self.input_size_per_partition, **factory))
        if bias:
            self.bias = nn.Parameter(torch.empty(output_size, **factory))
        else:
            self.register_parameter("bias", None)
        self.init_method = init_method
        self.reset_parameters()

    def reset_parameters(self):
        if self.weight.device.type == "meta":
            return
        init = self.init_method or (lambda w: nn.init.normal_(w, 0.0, 0.02))
        init(self.weight)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        ws = group_size(self.group)
        assert self.input_is_parallel
        out = torch.matmul(x, self.weight.t())
        if self.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out, self.group)
        elif ws > 1:
            out = reduce_from_tensor_model_parallel_region(out, self.group)
        if self.bias is not None:
            out = out + self.bias
        return out


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab dim sharded over the TP group.

    Reference: layers.py:59-147.  input_ids: [b, s]; output SBH [s, b, h];
    with sequence_parallel the output is reduce-scattered straight into the
    SP layout (layers.py:140-144).
    """

    def __init__(self, num_embeddings: int, embedding_dim: int, group,
                 sequence_parallel: bool = False, dtype=None,
                 init_method: Optional[Callable] = None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.group = group
        ws = group_size(group)
        rank = group_rank(group)
        self.vocab_size_per_partition = divide(num_embeddings, ws)
        self.vocab_start_index = rank * self.vocab_size_per_partition
        self.vocab_end_index = self.vocab_start_index + self.vocab_size_per_partition
        self.sequence_parallel = sequence_parallel and ws > 1
        factory = {"dtype": dtype} if dtype is not None else {}
        self.weight = nn.Parameter(torch.empty(
            self.vocab_size_per_partition, embedding_dim, **factory))
        self.init_method = init_method
        self.reset_parameters()

    def reset_parameters(self):
        if self.weight.device.type == "meta":
            return
        init = self.init_method or (lambda w: nn.init.normal_(w, 0.0, 0.02))
        init(self.weight)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        ws = group_size(self.group)
        if ws > 1:
            mask = (input_ids < self.vocab_start_index) | (input_ids >= self.vocab_end_index)
            masked = input_ids.clone() - self.vocab_start_index
            masked[mask] = 0
            out = F.embedding(masked, self.weight)
            out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        else:
            out = F.embedding(input_ids, self.weight)
        out = out.transpose(0, 1).contiguous()  # [b,s,h] -> [s,b,h]
        if self.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out, self.group)
        elif ws > 1:
            out = reduce_from_tensor_model_parallel_region(out, self.group)
        return out
