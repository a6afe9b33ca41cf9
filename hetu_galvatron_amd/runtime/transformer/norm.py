"""Norm layers over the CDNA4 fused kernels.

Reference: galvatron/core/runtime/transformer/norm.py:6-30 — there every
decoder-path norm is a flash-attn CUDA kernel (RMSNorm / DropoutAddLayerNorm);
here it is ops/csrc/rmsnorm.hip / layernorm.hip (torch reference on CPU).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import rms_norm, layer_norm


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5, dtype=None):
        super().__init__()
        factory = {"dtype": dtype} if dtype is not None else {}
        self.weight = nn.Parameter(torch.empty(hidden_size, **factory))
        self.eps = eps
        self.reset_parameters()

    def reset_parameters(self):
        if self.weight.device.type != "meta":
            nn.init.ones_(self.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, self.weight, self.eps)

    def forward_fused_add(self, x, res):
        """(normed, sum=x+res) in one kernel each way on the native path;
        eager equivalent elsewhere."""
        from ...ops._ext import use_native
        if use_native(x) and x.dtype == torch.bfloat16:
            from ...ops import fused_add_rms_norm
            return fused_add_rms_norm(x, res, self.weight, self.eps)
        s = x + res
        return rms_norm(s, self.weight, self.eps), s


class LayerNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5, dtype=None):
        super().__init__()
        factory = {"dtype": dtype} if dtype is not None else {}
        self.weight = nn.Parameter(torch.empty(hidden_size, **factory))
        self.bias = nn.Parameter(torch.empty(hidden_size, **factory))
        self.eps = eps
        self.reset_parameters()

    def reset_parameters(self):
        if self.weight.device.type != "meta":
            nn.init.ones_(self.weight)
            nn.init.zeros_(self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return layer_norm(x, self.weight, self.bias, self.eps)


def build_norm(normalization: str, hidden_size: int, eps: float, dtype=None) -> nn.Module:
    """Factory (reference: norm.py:6 GalvatronNorm)."""
    if normalization == "rmsnorm":
        return RMSNorm(hidden_size, eps, dtype)
    if normalization == "layernorm":
        return LayerNorm(hidden_size, eps, dtype)
    raise ValueError(f"unknown normalization {normalization}")
