"""MoE expert MLPs: grouped and sequential.

Reference: galvatron/core/runtime/moe/mlp.py:26-416 (GroupedMLP over
grouped_gemm.ops.gmm, SequentialMLP loop, SharedExpertMLP).

The grouped path batches each expert's variable-M token slab through
hipBLASLt GEMMs (torch.mm per expert on contiguous slices of the
expert-sorted buffer — on MI355X each slab GEMM is large enough that a
per-expert dispatch is GEMM-bound, not launch-bound); a fused CDNA4
grouped-GEMM kernel slots in here later without API change.
Expert parameters are tagged `expert_parallel` so the ZeRO engine reduces
their grads over the EDP group (reference parallel.py MoE double-wrap).
"""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from ...ops import swiglu
from ...ops.functional import grouped_gemm, grouped_gemm_available


def _mark_expert(p: nn.Parameter) -> nn.Parameter:
    p.expert_parallel = True
    return p


class GroupedMLP(nn.Module):
    """num_local_experts gated MLPs over an expert-sorted token buffer."""

    def __init__(self, num_local_experts: int, hidden_size: int,
                 ffn_hidden: int, dtype=None, gated: bool = True,
                 act: str = "silu", init_std: float = 0.02):
        super().__init__()
        self.num_local_experts = num_local_experts
        self.gated = gated
        self.act = act
        out1 = 2 * ffn_hidden if gated else ffn_hidden
        kw = {"dtype": dtype} if dtype else {}
        self.w1 = _mark_expert(nn.Parameter(
            torch.empty(num_local_experts, hidden_size, out1, **kw)))
        self.w2 = _mark_expert(nn.Parameter(
            torch.empty(num_local_experts, ffn_hidden, hidden_size, **kw)))
        nn.init.normal_(self.w1, 0.0, init_std)
        nn.init.normal_(self.w2, 0.0, init_std)

    def _act(self, h: torch.Tensor) -> torch.Tensor:
        import torch.nn.functional as F
        if not self.gated:
            return F.relu(h) if self.act == "relu" \
                else F.gelu(h, approximate="tanh")
        if self.act == "geglu":
            gate, up = h.chunk(2, dim=-1)
            return F.gelu(gate, approximate="tanh") * up
        return swiglu(h)

    def forward(self, x: torch.Tensor,
                tokens_per_expert: torch.Tensor) -> torch.Tensor:
        """x [m, h] expert-sorted; tokens_per_expert [E_local]."""
        sizes = [int(v) for v in tokens_per_expert]
        if grouped_gemm_available(x, self.w1) \
                and grouped_gemm_available(x, self.w2) and x.shape[0] > 0:
            h = grouped_gemm(x, self.w1, sizes)
            h = self._act(h)
            return grouped_gemm(h, self.w2, sizes)
        outs: List[torch.Tensor] = []
        start = 0
        for e, m in enumerate(sizes):
            xe = x[start:start + m]
            start += m
            if m == 0:
                continue  # unused experts: flat-grad segment stays zero
            h = xe @ self.w1[e]
            h = self._act(h)
            outs.append(h @ self.w2[e])
        return torch.cat(outs) if outs else x[:0]


class SequentialMLP(nn.Module):
    """Per-expert nn.Linear modules (reference moe/mlp.py:128)."""

    def __init__(self, num_local_experts: int, hidden_size: int,
                 ffn_hidden: int, dtype=None, gated: bool = True,
                 act: str = "silu", init_std: float = 0.02):
        super().__init__()
        self.gated = gated
        self.act = act
        out1 = 2 * ffn_hidden if gated else ffn_hidden
        kw = {"dtype": dtype} if dtype else {}
        self.fc1 = nn.ModuleList([
            nn.Linear(hidden_size, out1, bias=False, **kw)
            for _ in range(num_local_experts)])
        self.fc2 = nn.ModuleList([
            nn.Linear(ffn_hidden, hidden_size, bias=False, **kw)
            for _ in range(num_local_experts)])
        for m in list(self.fc1) + list(self.fc2):
            nn.init.normal_(m.weight, 0.0, init_std)
            _mark_expert(m.weight)

    def forward(self, x, tokens_per_expert):
        sizes = [int(v) for v in tokens_per_expert]
        outs, start = [], 0
        for e, m in enumerate(sizes):
            xe = x[start:start + m]
            start += m
            if m == 0:
                continue
            h = self.fc1[e](xe)
            h = GroupedMLP._act(self, h)
            outs.append(self.fc2[e](h))
        return torch.cat(outs) if outs else x[:0]


class SharedExpertMLP(nn.Module):
    """Always-on shared expert added to the routed output
    (reference moe/mlp.py:215)."""

    def __init__(self, hidden_size: int, inter: int, dtype=None,
                 init_std: float = 0.02):
        super().__init__()
        kw = {"dtype": dtype} if dtype else {}
        self.fc1 = nn.Linear(hidden_size, 2 * inter, bias=False, **kw)
        self.fc2 = nn.Linear(inter, hidden_size, bias=False, **kw)
        nn.init.normal_(self.fc1.weight, 0.0, init_std)
        nn.init.normal_(self.fc2.weight, 0.0, init_std)

    def forward(self, x):
        return self.fc2(swiglu(self.fc1(x)))
