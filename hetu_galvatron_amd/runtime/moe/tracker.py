"""Per-layer MoE aux-loss tracker with cross-rank reduction.

Reference: moe_utils.py:547-644 (save_to_aux_losses_tracker /
reduce_aux_losses_tracker_across_ranks / clear): routers record their
aux/z losses per layer each step; at logging time the values are
averaged over the dp group and exposed as a {name: value} dict.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist

_TRACKER: Dict[str, list] = {}


def save_aux_loss(name: str, layer_idx: int, value: torch.Tensor) -> None:
    _TRACKER.setdefault(f"{name}/layer_{layer_idx}", []).append(
        float(value.detach()) if torch.is_tensor(value) else float(value))


def reduce_and_get(group=None) -> Dict[str, float]:
    """Mean per key over recorded steps, then mean over the ranks that
    RECORDED the key.  Key sets differ across ranks under pipeline
    parallelism (each stage owns different MoE layers), so this gathers
    dicts (collective: every rank must call) instead of all-reducing a
    fixed-shape tensor — a shape-mismatched all_reduce would deadlock."""
    out = {k: sum(v) / max(len(v), 1) for k, v in _TRACKER.items()}
    if dist.is_initialized() and dist.get_world_size() > 1:
        ws = dist.get_world_size(group) if group is not None \
            else dist.get_world_size()
        gathered: list = [None] * ws
        dist.all_gather_object(gathered, out, group=group)
        merged: Dict[str, list] = {}
        for d in gathered:
            for k, v in (d or {}).items():
                merged.setdefault(k, []).append(v)
        out = {k: sum(v) / len(v) for k, v in merged.items()}
    return out


def clear() -> None:
    _TRACKER.clear()
