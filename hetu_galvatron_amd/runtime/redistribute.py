"""Activation redistribution between layers with different parallel layouts.

Reference: galvatron/core/runtime/redistribute.py:5-415 (zigzag transforms,
_Split/_Gather/_Fused_split_allgather).  Rebuilt from first principles with
an explicit canonical layout (the reference's helpers are convention-tangled
and its CP-only path is mis-wired — SURVEY §5):

Canonical microbatch activation: [S, B, h] (SBH).  A layout (tsp, cp, dp)
shards it as:
  batch : contiguous split into dp slices (dp_idx)
  seq   : zigzag split into cp chunk-pairs (cp_idx takes natural chunks
          (cp_idx, 2cp-1-cp_idx)), then the cp-local sequence is split into
          tsp contiguous slices (tp_idx)
so every rank holds [S/(cp*tsp), B/dp, h].

Redistribution A->B allgathers over the LARGER of the two tsp_cp groups
(power-of-2 consecutive layouts nest, so it contains the smaller), rebuilds
the natural-order subtensor owned by that group, and slices the target
layout.  Forward and backward are exact adjoints (each element has one owner
per layout).  The all-gather runs on RCCL over xGMI and is counted by the
cost model as the redistribution penalty (search_engine dynamic_programming
inter-cost).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..core.comm_groups import CommGroup, LayerCommGroups

_ROWS_CACHE: Dict[Tuple, torch.Tensor] = {}


def natural_rows(S: int, cp: int, tsp: int, cp_idx: int, tp_idx: int,
                 device) -> torch.Tensor:
    """Natural (global) seq row indices owned by (cp_idx, tp_idx)."""
    key = (S, cp, tsp, cp_idx, tp_idx, str(device))
    hit = _ROWS_CACHE.get(key)
    if hit is not None:
        return hit
    chunk = S // (2 * cp)
    lo = torch.arange(cp_idx * chunk, (cp_idx + 1) * chunk, device=device)
    hi = torch.arange((2 * cp - 1 - cp_idx) * chunk, (2 * cp - cp_idx) * chunk,
                      device=device)
    pair = torch.cat([lo, hi])  # cp-local zigzag order -> natural indices
    s_loc = (2 * chunk) // tsp
    rows = pair[tp_idx * s_loc:(tp_idx + 1) * s_loc].contiguous()
    _ROWS_CACHE[key] = rows
    return rows


def _same_seq_layout(a: LayerCommGroups, b: LayerCommGroups) -> bool:
    sa, sb = a.strategy, b.strategy
    return (sa.tp_sp == sb.tp_sp and sa.cp == sb.cp and sa.dp == sb.dp
            and a.tsp_cp_group.ranks == b.tsp_cp_group.ranks)


def _allgather_blocks(x: torch.Tensor, group: CommGroup) -> List[torch.Tensor]:
    if group.size == 1:
        return [x]
    out = [torch.empty_like(x) for _ in range(group.size)]
    dist.all_gather(out, x.contiguous(), group=group.group)
    return out


def _redistribute_impl(x: torch.Tensor, src: LayerCommGroups,
                       dst: LayerCommGroups, batch_global: int,
                       rank: int) -> torch.Tensor:
    """Move x from src layout shard to dst layout shard (no autograd)."""
    if _same_seq_layout(src, dst):
        return x
    ss, sd = src.strategy, dst.strategy
    # union group: the larger tsp_cp group (power-of-2 consecutive nesting)
    if src.tsp_cp_group.size >= dst.tsp_cp_group.size:
        U, big = src.tsp_cp_group, src
    else:
        U, big = dst.tsp_cp_group, dst
    assert set(src.tsp_cp_group.ranks) <= set(U.ranks) and \
        set(dst.tsp_cp_group.ranks) <= set(U.ranks), \
        f"layouts do not nest: {src.tsp_cp_group.ranks} / {dst.tsp_cp_group.ranks}"
    s_loc, b_loc = x.shape[0], x.shape[1]
    S = s_loc * ss.tp_sp * ss.cp
    # batch block owned by U = contiguous range covered by its members under src
    b_src = batch_global // ss.dp
    b_dst = batch_global // sd.dp
    src_batch_starts = {ru: src.coord_of(ru).dp_idx * b_src for ru in U.ranks}
    bu_start = min(src_batch_starts.values())
    B_U = b_src * len(set(src_batch_starts.values()))
    blocks = _allgather_blocks(x, U)
    full = torch.empty(S, B_U, *x.shape[2:], dtype=x.dtype, device=x.device)
    for i, ru in enumerate(U.ranks):
        c = src.coord_of(ru)
        rows = natural_rows(S, ss.cp, ss.tp_sp, c.cp_idx, c.tp_idx, x.device)
        bs = c.dp_idx * b_src - bu_start
        full[rows, bs:bs + b_src] = blocks[i]
    # slice own dst shard
    cd = dst.coord_of(rank)
    rows_d = natural_rows(S, sd.cp, sd.tp_sp, cd.cp_idx, cd.tp_idx, x.device)
    bs_d = cd.dp_idx * b_dst - bu_start
    assert 0 <= bs_d and bs_d + b_dst <= B_U, "dst batch slice outside union block"
    return full[rows_d, bs_d:bs_d + b_dst].contiguous()


class _Redistribute(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, src: LayerCommGroups, dst: LayerCommGroups,
                batch_global: int):
        rank = dist.get_rank() if dist.is_initialized() else 0
        ctx.src, ctx.dst, ctx.bg, ctx.rank = src, dst, batch_global, rank
        return _redistribute_impl(x, src, dst, batch_global, rank)

    @staticmethod
    def backward(ctx, dy):
        dx = _redistribute_impl(dy.contiguous(), ctx.dst, ctx.src, ctx.bg, ctx.rank)
        return dx, None, None, None


def redistribute(x: torch.Tensor, src: Optional[LayerCommGroups],
                 dst: Optional[LayerCommGroups], batch_global: int) -> torch.Tensor:
    """Re-shard activation x from src layout to dst layout (autograd-capable).
    reference: redistribute.py:408 fused_split_allgather."""
    if src is None or dst is None:
        return x
    if _same_seq_layout(src, dst):
        return x
    return _Redistribute.apply(x, src, dst, batch_global)
