"""Strategy-space enumeration + filtering + inter-layer transition costs.

Reference: galvatron/core/search_engine/search_engine.py:106-283
(generate_strategy_list / filter_strategy_list) and
dynamic_programming.py:161-210 (match_strategy redistribution penalty).
"""
from __future__ import annotations

from typing import List

from ..config.schema import SearchArgs
from ..config.strategy import LayerStrategy


def _pow2_range(limit: int) -> List[int]:
    out, v = [], 1
    while v <= limit:
        out.append(v)
        v *= 2
    return out


def enumerate_strategies(world_size: int, args: SearchArgs,
                         pp_deg: int,
                         default_dp_type: str = "ddp") -> List[LayerStrategy]:
    """All (tp|sp, dp_type, ckpt) combos for one pp degree.

    Power-of-2 degrees; tp*dp = world/pp; ulysses-sp layers replace tp by
    sp on the same degree (sp_space controls which are generated).
    """
    per_stage = world_size // pp_deg
    out: List[LayerStrategy] = []
    tp_degs = [t for t in _pow2_range(min(args.max_tp_deg, per_stage))]
    if args.disable_tp:
        tp_degs = [1]
    cp_degs = [1] if args.disable_cp else \
        _pow2_range(min(args.max_cp_deg, per_stage))
    for tsp in tp_degs:
        for cp in cp_degs:
            if tsp * cp > per_stage:
                continue
            dp = per_stage // (tsp * cp)
            if dp > 1 and args.disable_dp and tsp * cp * pp_deg != world_size:
                continue
            modes = []
            if tsp == 1:
                modes = [("tp", tsp)]
            else:
                if args.sp_space in ("tp", "tp+sp"):
                    modes.append(("tp", tsp))
                if args.sp_space in ("sp", "tp+sp") and not args.disable_sp \
                        and tsp <= getattr(args, "max_sp_deg", 8):
                    modes.append(("sp", tsp))
            for mode, deg in modes:
                tp = deg if mode == "tp" else 1
                sp = deg if mode == "sp" else 1
                # the "0" leg of dp_types_enc executes as the runtime's
                # default_dp_type — price what will actually run
                # the ZeRO domain for ulysses layers includes sp (params
                # replicated over sp; sdp = dp*cp*sp)
                sdp_deg = dp * cp * (sp if sp > 1 else 1)
                dp_types = [default_dp_type if sdp_deg > 1 else "ddp"]
                if not args.disable_sdp and sdp_deg > 1:
                    dp_types.append("zero3")
                for dpt in dp_types:
                    ckpts = [False] if args.disable_ckpt else [False, True]
                    for ck in ckpts:
                        out.append(LayerStrategy(
                            pp_deg=pp_deg, tp=tp, sp=sp, cp=cp, dp=dp,
                            dp_type=dpt, checkpoint=ck))
    return out


def strategy_key(s: LayerStrategy) -> str:
    return (f"pp{s.pp_deg}_tp{s.tp}_sp{s.sp}_cp{s.cp}_dp{s.dp}_"
            f"{s.dp_type}_ck{int(s.checkpoint)}")


def transition_cost_mb(prev: LayerStrategy, cur: LayerStrategy,
                       seq_len: int, hidden: int, local_bsz: float,
                       mixed_precision: bool = True) -> float:
    """Activation-redistribution penalty (MB moved) between two layouts.

    Reference dynamic_programming.py:161-210 keys the penalty on layout
    diffs (tp_sp/cp); here: if the (tp_sp, cp, use_sp) triple changes, the
    boundary costs an allgather of the full activation on the old group +
    a split — modelled as one activation's worth of bytes through the
    slower of the two layouts.
    """
    if (prev.tp_sp, prev.cp, prev.use_ulysses) == \
       (cur.tp_sp, cur.cp, cur.use_ulysses):
        return 0.0
    bytes_per = 2 if mixed_precision else 4
    act_mb = local_bsz * seq_len * hidden * bytes_per / (1024 * 1024)
    # allgather on the previous layout's group
    g = max(prev.tp_sp * prev.cp, cur.tp_sp * cur.cp)
    return act_mb * (g - 1) / g
