"""Communication-group fabric: rank coordinates -> RCCL process groups.

Reference: galvatron/core/runtime/comm_groups.py:5-442 (CommGroup,
CommGroupCache, build_rank_to_parallel_coords, gen_comm_groups).

MI355X-native design notes:
  * one process per GPU; groups are torch.distributed process groups over
    RCCL ("nccl" backend on ROCm) or gloo (CPU tests).
  * the 8-GPU xGMI mesh is fully connected (7 p2p links x ~153 GB/s/GPU), so
    there is no NVSwitch-style "consecutive is faster" asymmetry intra-node;
    the consec flag is still honored because it defines WHICH ranks share a
    group (the searched plans reference it) and because multi-node layouts
    reintroduce the asymmetry.
  * communicator count is bounded by caching on the exact rank tuple —
    layers that share a layout share communicators (RCCL communicators cost
    device memory; reference: CommGroupCache comm_groups.py:18-27).

Coordinate order within a pipeline stage (stage size G = world/pp):
  consecutive TP  : rank = (dp_idx*cp + cp_idx)*tsp + tp_idx   (tp fastest)
  non-consecutive : rank = (tp_idx*cp + cp_idx)*dp + dp_idx    (dp fastest)
This reproduces the reference's `pp-dp-cp-tp` order with sp folded into tp
(tp and ulysses-sp are mutually exclusive per layer).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import torch.distributed as dist

from ..config.strategy import LayerStrategy


class CommGroup:
    """A set of ranks + its (lazily created) torch.distributed process group."""

    def __init__(self, ranks: Sequence[int], group=None):
        self.ranks: Tuple[int, ...] = tuple(int(r) for r in ranks)
        self.group = group  # None for singleton groups (no comm needed)
        self.size = len(self.ranks)

    def contains(self, rank: int) -> bool:
        return rank in self.ranks

    def index(self, rank: int) -> int:
        return self.ranks.index(rank)

    def __repr__(self) -> str:
        return f"CommGroup(ranks={list(self.ranks)})"

    def __eq__(self, other) -> bool:
        return isinstance(other, CommGroup) and self.ranks == other.ranks

    def __hash__(self) -> int:
        return hash(self.ranks)


class CommGroupCache:
    """rank-tuple -> process group cache; creation is collective and must be
    invoked in identical order on every rank (reference: comm_groups.py:18-27)."""

    def __init__(self) -> None:
        self._cache: Dict[Tuple[int, ...], CommGroup] = {}

    def get(self, ranks: Sequence[int]) -> CommGroup:
        key = tuple(int(r) for r in ranks)
        if key in self._cache:
            return self._cache[key]
        if len(key) == 1 or not (dist.is_available() and dist.is_initialized()):
            cg = CommGroup(key, None)
        else:
            pg = dist.new_group(list(key))
            cg = CommGroup(key, pg)
        self._cache[key] = cg
        return cg

    def __len__(self) -> int:
        return len(self._cache)


@dataclass
class StageCoord:
    dp_idx: int
    cp_idx: int
    tp_idx: int


def build_stage_coords(G: int, tsp: int, cp: int, consecutive: bool) -> List[StageCoord]:
    """Coordinates of every in-stage rank r in [0,G) for layout (tsp, cp, dp).

    reference: comm_groups.py:39 build_rank_to_parallel_coords; docstring
    examples there are the executable spec our unit tests mirror.
    """
    assert G % (tsp * cp) == 0, f"stage size {G} not divisible by tsp*cp={tsp*cp}"
    dp = G // (tsp * cp)
    coords = []
    for r in range(G):
        if consecutive:
            tp_idx = r % tsp
            cp_idx = (r // tsp) % cp
            dp_idx = r // (tsp * cp)
        else:
            dp_idx = r % dp
            cp_idx = (r // dp) % cp
            tp_idx = r // (dp * cp)
        coords.append(StageCoord(dp_idx, cp_idx, tp_idx))
    return coords


@dataclass
class LayerCommGroups:
    """All groups one layer needs, from THIS rank's perspective."""

    strategy: LayerStrategy
    tp_group: CommGroup            # tensor-parallel (megatron TP / SP collectives)
    sp_group: CommGroup            # ulysses all-to-all group (== tp_group ranks)
    cp_group: CommGroup            # ring-attention context-parallel group
    dp_group: CommGroup            # pure data-parallel group
    sdp_group: CommGroup           # dp*cp — the ZeRO sharding / grad-reduce domain
    tsp_cp_group: CommGroup        # tp_sp*cp — sequence-layout domain (redistribution)
    stage_coords: Optional[List["StageCoord"]] = None  # in-stage rank -> coords
    stage_base: int = 0            # first global rank of this pp stage
    # MoE (None for dense layers)
    ep_group: Optional[CommGroup] = None      # expert-parallel all-to-all
    edp_group: Optional[CommGroup] = None     # data-parallel of experts (grad reduce)
    etp_group: Optional[CommGroup] = None     # tp within experts

    def coord_of(self, global_rank: int) -> "StageCoord":
        return self.stage_coords[global_rank - self.stage_base]

    @property
    def seq_shard_degree(self) -> int:
        """How many ways the sequence dim is sharded between layers."""
        s = self.strategy
        deg = s.cp
        if s.use_ulysses or (s.tp > 1):
            deg *= s.tp_sp
        return deg


def _groups_for_layout(
    world_size: int, pp_deg: int, tsp: int, cp: int, consecutive: bool,
    cache: CommGroupCache, rank: int, ep: int = 1, ulysses: bool = False,
) -> Dict[str, CommGroup]:
    """Create (collectively) all groups of one layout; return this rank's."""
    G = world_size // pp_deg
    dp = G // (tsp * cp)
    coords = build_stage_coords(G, tsp, cp, consecutive)
    mine: Dict[str, CommGroup] = {}

    def make(sel_fn, name: str) -> None:
        # enumerate groups deterministically over all stages and key slots
        for stage in range(pp_deg):
            base = stage * G
            buckets: Dict[Tuple, List[int]] = {}
            for r, c in enumerate(coords):
                key = sel_fn(c)
                buckets.setdefault(key, []).append(base + r)
            for key in sorted(buckets.keys()):
                ranks = buckets[key]
                cg = cache.get(ranks)
                if rank in ranks:
                    mine[name] = cg

    make(lambda c: (c.dp_idx, c.cp_idx), "tp")
    make(lambda c: (c.dp_idx, c.tp_idx), "cp")
    make(lambda c: (c.cp_idx, c.tp_idx), "dp")
    # ZeRO / grad-reduce domain = all ranks holding identical param copies:
    # megatron-TP shards params over tp -> sdp = dp x cp; ulysses replicates
    # them over sp too -> sdp = dp x cp x sp (reference: comm_groups.py:310,
    # SDP group merges dp/sp/cp).
    if ulysses:
        make(lambda c: (), "sdp")
    else:
        make(lambda c: (c.tp_idx,), "sdp")
    make(lambda c: (c.dp_idx,), "tsp_cp")

    if ep > 1:
        # MoE coordinate family (pp, ep, edp, etp): experts sharded over ep
        # within the sdp domain; etp == tsp (reference: comm_groups.py:322-345).
        sdp = dp * cp
        assert sdp % ep == 0, f"ep={ep} must divide sdp={sdp}"
        # flatten (dp_idx, cp_idx) -> sdp_idx; split into (edp_idx, ep_idx), ep fastest
        def sdp_idx(c):
            return c.dp_idx * cp + c.cp_idx
        make(lambda c: (c.tp_idx, sdp_idx(c) // ep), "ep")    # vary ep_idx
        make(lambda c: (c.tp_idx, sdp_idx(c) % ep), "edp")    # vary edp_idx
        mine["etp"] = mine["tp"]
    return mine


def gen_layer_comm_groups(
    strategies: Sequence[LayerStrategy], world_size: int, rank: int,
    cache: Optional[CommGroupCache] = None,
) -> Tuple[List[LayerCommGroups], CommGroupCache]:
    """Create comm groups for every layer of a plan.

    Collective: every rank calls with identical `strategies` and enumerates
    identical group lists in identical order (reference: gen_comm_groups
    comm_groups.py:266).
    """
    cache = cache or CommGroupCache()
    out: List[LayerCommGroups] = []
    for s in strategies:
        layout = _groups_for_layout(
            world_size, s.pp_deg, s.tp_sp, s.cp, s.tp_consecutive, cache, rank,
            ep=s.ep, ulysses=s.use_ulysses,
        )
        G = world_size // s.pp_deg
        out.append(LayerCommGroups(
            strategy=s,
            tp_group=layout["tp"], sp_group=layout["tp"], cp_group=layout["cp"],
            dp_group=layout["dp"], sdp_group=layout["sdp"],
            tsp_cp_group=layout["tsp_cp"],
            stage_coords=build_stage_coords(G, s.tp_sp, s.cp, s.tp_consecutive),
            stage_base=(rank // G) * G,
            ep_group=layout.get("ep"), edp_group=layout.get("edp"),
            etp_group=layout.get("etp"),
        ))
    return out, cache


def gen_embedding_group(world_size: int, pp_deg: int, cache: CommGroupCache,
                        rank: int) -> Optional[CommGroup]:
    """Group tying first-stage embedding and last-stage lm-head gradients
    (reference: comm_groups.py:206 get_embedding_group)."""
    if pp_deg == 1:
        return None
    G = world_size // pp_deg
    mine = None
    for i in range(G):
        ranks = [i, (pp_deg - 1) * G + i]
        cg = cache.get(ranks)
        if rank in ranks:
            mine = cg
    return mine


def pp_stage_of_rank(rank: int, world_size: int, pp_deg: int) -> int:
    return rank // (world_size // pp_deg)


def pp_neighbor_ranks(rank: int, world_size: int, pp_deg: int) -> Tuple[Optional[int], Optional[int]]:
    """(prev, next) global ranks holding the same in-stage index."""
    G = world_size // pp_deg
    stage = rank // G
    idx = rank % G
    prev_rank = (stage - 1) * G + idx if stage > 0 else None
    next_rank = (stage + 1) * G + idx if stage < pp_deg - 1 else None
    return prev_rank, next_rank


def describe_groups(layer_groups: Sequence[LayerCommGroups]) -> str:
    """Pretty printer (reference: comm_groups.py:385-422)."""
    lines = []
    for i, g in enumerate(layer_groups):
        s = g.strategy
        lines.append(
            f"layer {i}: tp={s.tp} sp={s.sp} cp={s.cp} dp={s.dp} ({s.dp_type})"
            f" ckpt={int(s.checkpoint)} | tp{list(g.tp_group.ranks)}"
            f" cp{list(g.cp_group.ranks)} sdp{list(g.sdp_group.ranks)}")
    return "\n".join(lines)
