"""Transformer MLP: ColumnParallel fc1 -> fused activation -> RowParallel fc2.

Reference: galvatron/core/runtime/transformer/mlp.py:23-133 with the fused
bias-activation kernels of transformer/fused_kernels.py:101-226.  The SwiGLU
elementwise pass is the HIP kernel ops/csrc/swiglu.hip (fused silu*up in one
HBM round trip — the op is bandwidth-bound on MI355X's 8 TB/s HBM3E).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import swiglu
from ..tensor_parallel import ColumnParallelLinear, RowParallelLinear


class MLP(nn.Module):
    def __init__(self, hidden_size: int, ffn_hidden_size: int, group,
                 hidden_act: str = "silu", add_bias: bool = False,
                 sequence_parallel: bool = False, dtype=None,
                 init_method=None, output_init_method=None):
        super().__init__()
        self.hidden_act = hidden_act
        self.gated = hidden_act in ("silu", "swiglu", "geglu")
        fc1_out = ffn_hidden_size * (2 if self.gated else 1)
        self.fc1 = ColumnParallelLinear(
            hidden_size, fc1_out, group, bias=add_bias,
            sequence_parallel=sequence_parallel, dtype=dtype,
            init_method=init_method)
        self.fc2 = RowParallelLinear(
            ffn_hidden_size, hidden_size, group, bias=add_bias,
            sequence_parallel=sequence_parallel, dtype=dtype,
            init_method=output_init_method)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.fc1(x)
        if self.hidden_act in ("silu", "swiglu"):
            h = swiglu(h)
        elif self.hidden_act == "geglu":
            gate, up = h.chunk(2, dim=-1)
            h = F.gelu(gate, approximate="tanh") * up
        elif self.hidden_act == "gelu":
            h = F.gelu(h, approximate="tanh")
        elif self.hidden_act == "relu":
            h = F.relu(h)
        else:
            raise ValueError(f"unknown activation {self.hidden_act}")
        return self.fc2(h)
