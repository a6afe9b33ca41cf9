"""Analytic per-layer time/memory cost models + pipeline makespan.

Reference behavior: galvatron/core/cost_model/components/layer_cost.py:9-328
(TimeCostModelBase / MemoryCostModelBase), embedding_lmhead_cost.py,
cost_model_handler.py:16-99 (pipeline_costmodel).  Re-designed as plain
functions over two small profile bundles instead of the reference's
five-arg-object classes; formulas are kept semantically equivalent so the
search results transfer:

  time:  fct = linear_fit(local_bsz / tp_sp) ; bct = coe*fct (+fct if ckpt)
         dp comm = 2(d-1)/d * param_MB (halved for bf16) at profiled
         per-MB allreduce latency (consec keyed), overlap-modelled against
         bct; tp comm = 6 allgathers (megatron-sp) or 4 a2a (ulysses) at
         profiled message-size latency (x1.5 if ckpt); zero3 adds the fwd
         allgather; pp adds the p2p term.
  memory: model states = 4*param/tp * zero2/zero3 ratio (mixed-precision
         aware, +0.003 shard pad); activation = profiled per-bsz dict *
         cumulative microbatches (1F1B warmup depth pp-stage aware).
  pipeline: sum(stage_compute) + last_stage*(chunks-1), warmup/cooldown
         bubble lower bound with 1/3+2/3 split, non-overlapped grad-reduce
         remainder.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from ..config.strategy import LayerStrategy


@dataclass
class LayerProfile:
    """Per-layer-type profiled numbers (one decoder layer)."""

    parameter_mb: float                  # fp32-equivalent parameter MB
    fct_linear: Tuple[float, float]      # fwd ms = m * local_bsz + c (seq fixed)
    act_per_bsz_mb: Dict[str, float]     # {"1","2","4","8","checkpoint"} -> MB per sample
    seq_length: int
    hidden_size: int


@dataclass
class OtherProfile:
    """Embedding + final-norm + lm-head ("other") costs."""

    parameter_mb: float = 0.0            # embedding+head param MB (pre-vtp split)
    act_per_bsz_mb: Dict[str, float] = field(default_factory=dict)
    fct_linear: Tuple[float, float] = (0.0, 0.0)


@dataclass
class HardwareProfile:
    """Profiled RCCL-over-xGMI communication behavior (§2.2 outputs)."""

    # per-MB allreduce latency (ms/MB) keyed "{size}_{consec}"
    allreduce_latency_per_mb: Dict[str, float] = field(default_factory=dict)
    # message-size (MB) -> latency (ms) dicts per group size, with "popt" fit
    allgather_latency: Dict[int, Dict] = field(default_factory=dict)
    all2all_latency: Dict[int, Dict] = field(default_factory=dict)
    # p2p per-MB latency coefficient per pp_size (ms/MB)
    p2p_latency_per_mb: Dict[int, float] = field(default_factory=dict)
    overlap_coe: float = 1.15
    bct_fct_coe: float = 2.0
    extra_overhead: float = 0.0
    costmodel_coe: float = 1.0


def _msg_latency(select: Dict, mb: float) -> float:
    if mb in select:
        return select[mb]
    key = str(mb)
    if key in select:
        return select[key]
    m, c = select["popt"]
    return m * mb + c


def ddp_ratio(mixed_precision: bool) -> float:
    """Model-states bytes relative to 16 B/param, for THIS runtime's flat
    engine (zero.py): bf16 param 2 + fp32 grad accumulator 4 (FULL — grads
    accumulate locally across microbatches before reduction) + fp32 master
    4 + Adam moments 8."""
    return 18 / 16 if mixed_precision else 1.0


def zero2_ratio(d: int, mixed_precision: bool,
                grad_accum_in_shard: bool = False) -> float:
    """bf16 param full 2 + fp32 grad full 4 + reduced grad shard 4/d +
    (master 4 + moments 8)/d.  grad_accum_in_shard
    (parallel.reduce_grads_each_microbatch): the full fp32 accumulator is
    replaced by per-microbatch reduce-scatters into the shard."""
    if not mixed_precision:
        return 3 / 4 * (1 / d + 0.003) + 1 / 4
    if d <= 1:
        return ddp_ratio(mixed_precision)
    if grad_accum_in_shard:
        return 2 / 16 + (1 / d + 0.003)
    return 6 / 16 + (1 / d + 0.003)


def zero3_ratio(d: int, mixed_precision: bool,
                grad_accum_in_shard: bool = False) -> float:
    """bf16 shard 2/d (+ transient gathered block, amortized into the pad)
    + fp32 grad full 4 + grad shard 4/d + masters/moments 12/d; with
    grad_accum_in_shard the full fp32 term drops (one in-flight block's
    wire buffer amortized into the pad)."""
    if not mixed_precision:
        return 1 / d + 0.003
    if grad_accum_in_shard:
        return (18 / 16) * (1 / d + 0.003) + 0.02
    return 4 / 16 + (18 / 16) * (1 / d + 0.003)


def layer_time_cost(s: LayerStrategy, prof: LayerProfile, hw: HardwareProfile,
                    global_bsz: int, chunks: int, world_size: int,
                    mixed_precision: bool = True,
                    no_gradient_sync: bool = False) -> float:
    """One layer's per-microbatch time (ms), reference layer_cost.py:9-213."""
    lbsz = global_bsz / chunks / s.dp
    m, c = prof.fct_linear
    fct = m * (lbsz / (s.tp_sp * s.cp)) + c
    bct = fct * hw.bct_fct_coe
    if s.checkpoint:
        bct += fct

    # -- DP gradient sync --------------------------------------------------
    # unit = one half-collective (reduce-scatter OR all-gather) of the
    # layer's wire-dtype grads/params over the sdp group.  Mode split
    # (matches zero.py's schedule):
    #   ddp   - allreduce during backward: 2 units overlappable
    #   zero2 - RS during backward (1 unit overlappable) + param AG after
    #           the optimizer step (1 unit exposed)
    #   zero3 - bwd param re-gather + grad RS (2 units overlappable) +
    #           fwd param AG (1 unit exposed; forward has no overlap)
    sdp = s.sdp
    unit_mb = (sdp - 1) / max(sdp, 1) * (prof.parameter_mb / max(s.tp, 1))
    if mixed_precision:
        unit_mb /= 2
    if no_gradient_sync:
        unit_mb = 0.0
    key = f"{sdp}_0" if s.tp > 1 else f"{sdp}_1"
    dc = hw.allreduce_latency_per_mb.get(key, 0.0) if sdp > 1 else 0.0
    dc_overlap = dc * hw.overlap_coe
    if s.dp_type == "zero3":
        dp_message_mb, exposed_mb = 2 * unit_mb, unit_mb
    elif s.dp_type == "zero2":
        dp_message_mb, exposed_mb = unit_mb, unit_mb
    else:
        dp_message_mb, exposed_mb = 2 * unit_mb, 0.0

    # -- TP / SP collectives ----------------------------------------------
    tp_time = 0.0
    if s.tp_sp > 1:
        msg_mb = (lbsz * prof.seq_length * prof.hidden_size *
                  (2 if mixed_precision else 4) / (1024 * 1024)) / s.cp
        if s.use_ulysses:
            n_comm = 4
            select = hw.all2all_latency.get(s.sp)
        else:
            n_comm = 6
            select = hw.allgather_latency.get(s.tp)
        if s.checkpoint:
            n_comm = int(n_comm * 1.5)
        if select:
            tp_time = _msg_latency(select, msg_mb) * n_comm

    # -- ring-CP kv exchange ----------------------------------------------
    # zigzag ring: fwd rotates (k, v) cp-1 steps, bwd rotates (k, v, dk,
    # dv) cp steps; per-step message = this rank's kv shard.  Priced at
    # the 2-party p2p rate (dedicated xGMI links, overlappable with the
    # attention MFMA work -> overlap_coe discount).
    cp_time = 0.0
    if s.cp > 1:
        kv_mb = (lbsz * prof.seq_length / s.cp * prof.hidden_size *
                 (2 if mixed_precision else 4) / (1024 * 1024))
        p2p = hw.p2p_latency_per_mb.get(2, 0.0)
        cp_time = kv_mb * p2p * (2 * (s.cp - 1) + 4 * s.cp) / hw.overlap_coe

    # -- combine with backward/DP overlap model ---------------------------
    if sdp > 1 and dc > 0:
        dp_overlap_time = dp_message_mb * dc_overlap
        bct_overlap_time = bct * hw.overlap_coe
        if dp_overlap_time > bct_overlap_time:
            overlap_part = bct_overlap_time
            rest_part = (dp_message_mb - bct_overlap_time / dc_overlap) * dc
        else:
            overlap_part = dp_overlap_time
            rest_part = bct - dp_overlap_time / hw.overlap_coe
        result = fct + overlap_part + rest_part + tp_time + cp_time \
            + hw.extra_overhead
    else:
        result = fct + bct + tp_time + cp_time

    # -- exposed (non-overlappable) DP traffic ----------------------------
    if sdp > 1:
        result += exposed_mb * dc

    return result * hw.costmodel_coe  # ms


def layer_p2p_cost(s: LayerStrategy, prof: LayerProfile, hw: HardwareProfile,
                   global_bsz: int, chunks: int,
                   mixed_precision: bool = True) -> float:
    """Per-stage-boundary p2p time (added once per stage, not per layer)."""
    if s.pp_deg <= 1 or s.pp_deg not in hw.p2p_latency_per_mb:
        return 0.0
    lbsz = global_bsz / chunks / s.dp
    mb = (2 * lbsz * prof.seq_length * prof.hidden_size * 4 / (1024 * 1024))
    if mixed_precision:
        mb /= 2
    return mb * hw.p2p_latency_per_mb[s.pp_deg]


def layer_memory_cost(s: LayerStrategy, prof: LayerProfile,
                      global_bsz: int, chunks: int, stage_idx: int,
                      pipeline_type: str = "pipedream_flush",
                      mixed_precision: bool = True,
                      grad_accum_in_shard: bool = False) -> Dict[str, float]:
    """One layer's memory (MB): model states + activation.
    Reference layer_cost.py:215-328."""
    lbsz = global_bsz / chunks / s.dp
    if s.pp_deg == 1:
        cumulative = 1
    elif pipeline_type == "pipedream_flush":
        cumulative = s.pp_deg - stage_idx
    else:  # gpipe holds every microbatch's activations
        cumulative = chunks
    cum_lbsz = cumulative * lbsz

    # params shard over megatron-tp only; ulysses keeps them whole and the
    # ZeRO ratio (over sdp = dp*cp*sp) does the sharding
    param_mb = prof.parameter_mb / max(s.tp, 1)
    states = 4 * param_mb
    if s.dp_type == "zero3":
        states *= zero3_ratio(s.sdp, mixed_precision, grad_accum_in_shard)
    elif s.dp_type == "zero2":
        states *= zero2_ratio(s.sdp, mixed_precision, grad_accum_in_shard)
    else:
        states *= ddp_ratio(mixed_precision)

    if s.checkpoint:
        act = prof.act_per_bsz_mb["checkpoint"] * cum_lbsz / s.cp
        act /= s.tp_sp
    else:
        d = prof.act_per_bsz_mb
        per_bsz = d.get(str(s.tp_sp), d.get(s.tp_sp))
        if per_bsz is None:
            # profile measured at tp=1 only (1-GPU lease): activations
            # shard ~1/tp_sp under Megatron-SP / Ulysses
            per_bsz = d.get("1", d.get(1)) / s.tp_sp
        act = per_bsz * cum_lbsz / s.cp
    return {"parameter": param_mb, "model_states": states, "activation": act,
            "total": states + act}


def stage_sums(layer_costs: Sequence[float], division: Sequence[int]) -> List[float]:
    assert sum(division) == len(layer_costs)
    out, i = [], 0
    for n in division:
        out.append(float(sum(layer_costs[i:i + n])))
        i += n
    return out


def pipeline_cost(stage_compute: List[float], stage_with_sync: List[float],
                  chunks: int, pp: int,
                  other_time_cost: Optional[List[float]] = None) -> float:
    """1F1B makespan approximation (reference cost_model_handler.py:16-99)."""
    comp = list(stage_compute)
    if other_time_cost:
        comp = [c + o for c, o in zip(comp, other_time_cost)]
    result = sum(comp) + comp[-1] * (chunks - 1)
    if pp > 1:
        warm = min(pp - 1, chunks - 1)
        bubble = (max(warm * comp[0] * (1 / 3), sum(comp[1:]) * (1 / 3)) +
                  max(warm * comp[0] * (2 / 3), sum(comp[1:]) * (2 / 3)) +
                  comp[0] * max(0, chunks + 1 - pp))
        result = max(result, bubble)
    # non-overlapped gradient-reduce remainder
    reduce_rest = [stage_with_sync[i] - sum(comp[:i + 1]) for i in range(pp)]
    rest = max(reduce_rest) if reduce_rest else 0.0
    return result + max(rest, 0.0)
