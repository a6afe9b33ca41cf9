"""Layer-wise hybrid-parallel strategy search engine.

Reference: galvatron/core/search_engine/search_engine.py:21-1099
(GalvatronSearchEngine) + dynamic_programming.py:117-648 (DpOnModel).
Re-designed flow, same inputs/outputs:

  profiled JSONs (computation/memory/hardware, reference schemas)
      -> curve fits -> LayerProfile / OtherProfile / HardwareProfile
  task grid {global_bsz x chunks x pp_deg}
      -> per pp stage: strategy enumeration + DP (C++ core) under the
         per-GPU memory budget, inter-layer transition costs
      -> 1F1B makespan -> throughput; best task's per-layer plan is saved
         as the searched-config JSON (config.strategy codec — the
         search<->runtime contract).

Unlike the reference (single DP + pp-division heuristics), pp>1 solves a
DP per stage with the stage's own activation-accumulation depth, which is
both simpler and stage-exact for 1F1B.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..config import GalvatronConfig
from ..config.strategy import HybridParallelPlan, LayerStrategy, write_json_config
from .costmodel import (
    HardwareProfile,
    LayerProfile,
    OtherProfile,
    layer_memory_cost,
    layer_p2p_cost,
    layer_time_cost,
    pipeline_cost,
)
from .dp import backtrace, solve_layer_dp
from .strategies import enumerate_strategies, transition_cost_mb


def fit_linear(xs: List[float], ys: List[float]) -> Tuple[float, float]:
    m, c = np.polyfit(np.asarray(xs, dtype=float),
                      np.asarray(ys, dtype=float), 1)
    return float(m), float(c)


def read_computation_profile(path_or_dict, seq_len: int) -> Tuple[float, float]:
    """Reference computation_profiling JSON -> per-layer fwd ms linear fit.

    Keys "layernum[N]_bsz{b}_seq{s}": whole-model fwd ms.  Per-layer time
    by layernum differencing (reference model_profiler.py:422-472):
    t_layer(b) = (t[N2](b) - t[N1](b)) / (N2 - N1).
    """
    d = path_or_dict
    if isinstance(d, str):
        with open(d) as f:
            d = json.load(f)
    import re
    by_ln: Dict[int, Dict[int, float]] = {}
    for k, v in d.items():
        m = re.match(r"layernum\[(\d+)\]_bsz(\d+)_seq(\d+)", k)
        if not m or int(m.group(3)) != seq_len:
            continue
        by_ln.setdefault(int(m.group(1)), {})[int(m.group(2))] = float(v)
    lns = sorted(by_ln)
    assert len(lns) >= 2, f"need >=2 layernum sweeps, got {lns}"
    n1, n2 = lns[0], lns[-1]
    xs, ys = [], []
    for b in sorted(set(by_ln[n1]) & set(by_ln[n2])):
        xs.append(b)
        ys.append((by_ln[n2][b] - by_ln[n1][b]) / (n2 - n1))
    m, c = fit_linear(xs, ys)
    # clamp the intercept: a small negative fit residue would make the DP
    # chunk term (chunks * c) reward infinite microbatching
    return (m, max(c, 0.0))


def read_other_computation_profile(path_or_dict, seq_len: int
                                   ) -> Tuple[float, float]:
    """Embedding+head+loss ("other") fwd ms from the SAME layernum sweep:
    t_other(b) = t[N1](b) - N1 * t_layer(b) — the whole-model time minus
    the extrapolated decoder-layer share (includes fixed launch overhead,
    which honestly belongs to the non-repeating part)."""
    d = path_or_dict
    if isinstance(d, str):
        with open(d) as f:
            d = json.load(f)
    import re
    by_ln: Dict[int, Dict[int, float]] = {}
    for k, v in d.items():
        m = re.match(r"layernum\[(\d+)\]_bsz(\d+)_seq(\d+)", k)
        if not m or int(m.group(3)) != seq_len:
            continue
        by_ln.setdefault(int(m.group(1)), {})[int(m.group(2))] = float(v)
    lns = sorted(by_ln)
    assert len(lns) >= 2, f"need >=2 layernum sweeps, got {lns}"
    n1, n2 = lns[0], lns[-1]
    xs, ys = [], []
    for b in sorted(set(by_ln[n1]) & set(by_ln[n2])):
        t_layer = (by_ln[n2][b] - by_ln[n1][b]) / (n2 - n1)
        xs.append(b)
        ys.append(max(by_ln[n1][b] - n1 * t_layer, 0.0))
    m, c = fit_linear(xs, ys)
    return (m, max(c, 0.0))


def read_hardware_profiles(hw_dir: str, nodes: int = 1, gpus: int = 8
                           ) -> HardwareProfile:
    """Reference hardware_configs/*.json -> HardwareProfile."""
    hw = HardwareProfile()
    suff = f"{nodes}nodes_{gpus}gpus_per_node.json"

    def load(name):
        p = os.path.join(hw_dir, f"{name}_{suff}")
        if os.path.exists(p):
            with open(p) as f:
                return json.load(f)
        return {}

    ar = load("allreduce_bandwidth")
    for k, v in ar.items():  # "allreduce_size_8_consec_1": GB/s
        parts = k.split("_")
        size, consec = parts[2], parts[4]
        # per-MB latency in ms from bus bandwidth: t = 2(n-1)/n * MB / BW
        n = int(size)
        hw.allreduce_latency_per_mb[f"{size}_{consec}"] = \
            1.0 / (float(v) * 1024) * 1000 * 2 * (n - 1) / n
    p2p = load("p2p_bandwidth")
    for k, v in p2p.items():  # "pp_size_2": GB/s
        hw.p2p_latency_per_mb[int(k.split("_")[-1])] = \
            1.0 / (float(v) * 1024) * 1000
    sp = load("sp_time")
    for k, v in sp.items():  # "allreduce_size_8_16MB_time" / "all2all_..."
        parts = k.split("_")
        op, size, mb = parts[0], int(parts[2]), float(parts[3][:-2])
        tgt = hw.allgather_latency if op == "allreduce" else hw.all2all_latency
        d = tgt.setdefault(size, {})
        # allgather/RS each ~half an allreduce of the same payload
        d[mb] = float(v) / (2.0 if op == "allreduce" else 1.0)
    for tgt in (hw.allgather_latency, hw.all2all_latency):
        for size, d in tgt.items():
            pts = sorted((k, v) for k, v in d.items() if isinstance(k, float))
            if len(pts) >= 2:
                d["popt"] = fit_linear([p[0] for p in pts], [p[1] for p in pts])
            elif pts:
                d["popt"] = (0.0, pts[0][1])
    ov_path = os.path.join(hw_dir, "overlap_coefficient.json")
    if os.path.exists(ov_path):
        with open(ov_path) as f:
            hw.overlap_coe = float(json.load(f).get("overlap_coe", 1.15))
    return hw


@dataclass
class SearchResult:
    throughput: float            # samples/s
    plan: HybridParallelPlan
    time_ms: float
    global_bsz: int
    chunks: int
    pp_deg: int
    vtp: int
    memory_mb: List[float] = field(default_factory=list)


class SearchEngine:
    """Auto-parallel plan search (single process, CPU)."""

    def __init__(self, cfg: GalvatronConfig,
                 layer_profile: Optional[LayerProfile] = None,
                 other_profile: Optional[OtherProfile] = None,
                 hardware: Optional[HardwareProfile] = None,
                 mem_unit_mb: int = 64):
        self.cfg = cfg
        self.args = cfg.search
        self.world = self.args.num_nodes * self.args.num_gpus_per_node
        m = cfg.model
        if m.model_type == "t5":
            n_enc = m.num_hidden_layers
            n_dec = m.num_decoder_layers or n_enc
            self.num_layers = n_enc + n_dec
        else:
            self.num_layers = m.num_hidden_layers
        self.layer_profile = layer_profile
        self.other_profile = other_profile or OtherProfile()
        self.hw = hardware
        coe = float(getattr(self.args, "debug_costmodel_coe", 1.0))
        if self.hw is not None and coe != 1.0:
            from dataclasses import replace
            self.hw = replace(self.hw, costmodel_coe=self.hw.costmodel_coe * coe)
        self.mem_unit = mem_unit_mb
        self.mixed_precision = cfg.parallel.mixed_precision == "bf16"
        self.results: List[SearchResult] = []

    @property
    def layer_profiles(self) -> List[LayerProfile]:
        """Per-layer-type profiles (reference: multi-layer-type DP,
        dynamic_programming.py _build_dp_and_run_multi_layer_type).
        `layer_profile` may be a single LayerProfile or a list."""
        lp = self.layer_profile
        return list(lp) if isinstance(lp, (list, tuple)) else [lp]

    @property
    def layer_types(self) -> List[int]:
        """Layer index -> profile index.  t5: encoder layers type 0,
        decoder layers type 1 when two profiles are given."""
        m = self.cfg.model
        n_types = len(self.layer_profiles)
        if m.model_type == "t5" and n_types > 1:
            n_enc = m.num_hidden_layers
            return [0] * n_enc + [1] * (self.num_layers - n_enc)
        return [0] * self.num_layers

    # -- profile loading ----------------------------------------------------
    def load_profiles(self, comp_path: str, mem_path: str, hw_dir: str) -> None:
        seq = self.cfg.model.seq_length
        fct = read_computation_profile(comp_path, seq)
        with open(mem_path) as f:
            mem = json.load(f)
        profiles = []
        for t in range(8):
            lt = mem.get(f"layertype_{t}")
            if lt is None:
                break
            profiles.append(LayerProfile(
                parameter_mb=float(lt["parameter_size"]),
                fct_linear=tuple(lt.get("fct_linear", fct)),
                act_per_bsz_mb=lt["tp_activation_per_bsz_dict"],
                seq_length=seq, hidden_size=self.cfg.model.hidden_size))
        assert profiles, "memory profile JSON missing layertype_0"
        self.layer_profile = profiles[0] if len(profiles) == 1 else profiles
        other = mem.get("other", {})
        self.other_profile = OtherProfile(
            parameter_mb=float(other.get("parameter_size", 0.0)),
            act_per_bsz_mb=other.get("tp_activation_per_bsz_dict", {}),
            fct_linear=tuple(other.get(
                "fct_linear",
                read_other_computation_profile(comp_path, seq))))
        self.hw = read_hardware_profiles(hw_dir, self.args.num_nodes,
                                         self.args.num_gpus_per_node)

    # -- task grid ----------------------------------------------------------
    def _bsz_candidates(self) -> List[int]:
        a = self.args
        if a.settle_bsz and a.settle_bsz > 0:
            return [a.settle_bsz]
        lo = a.min_bsz
        if getattr(a, "recommend_min_bsz", 0):
            # start where every GPU has at least one sample per chunk
            lo = max(lo, self.world)
        out, b = [], lo
        while b <= a.max_bsz:
            out.append(b)
            b += a.bsz_scale
        return out

    def _chunk_candidates(self, bsz: int, pp: int) -> List[int]:
        if self.args.settle_chunks and self.args.settle_chunks > 0:
            return [self.args.settle_chunks]
        cands = sorted({c for c in (1, 2, 4, 8, 16)
                        if c <= bsz and (pp == 1 or c >= pp)})
        return cands or [max(pp, 1)]

    def _pp_candidates(self) -> List[int]:
        a = self.args
        out, p = [], 1
        while p <= min(a.max_pp_deg, self.world, self.num_layers):
            if not (a.disable_pp and p > 1):
                out.append(p)
            p *= 2
        return out

    # -- core search --------------------------------------------------------
    def search_task(self, global_bsz: int, chunks: int, pp: int
                    ) -> Optional[SearchResult]:
        profiles, hw = self.layer_profiles, self.hw
        lp = profiles[0]
        ltypes = self.layer_types
        strategies = enumerate_strategies(
            self.world, self.args, pp,
            default_dp_type=self.cfg.parallel.default_dp_type)
        strategies = [s for s in strategies
                      if global_bsz % (s.dp * chunks) == 0]
        if not strategies:
            return None
        S = len(strategies)

        budget_mb = self.args.memory_constraint * 1024.0
        budget_mb -= 2048.0  # allocator/runtime context reserve

        # "other" (embedding/head) per vocab-tp choice
        vtp_opts = [1] if self.args.disable_vtp else \
            [v for v in (1, 2, 4, 8) if v <= self.world // pp]
        other_mem = {}
        for vtp in vtp_opts:
            pmb = self.other_profile.parameter_mb / vtp
            states = 4 * pmb
            act = self.other_profile.act_per_bsz_mb.get(
                str(vtp), self.other_profile.act_per_bsz_mb.get("1", 0.0))
            act = float(act) * global_bsz / chunks / max(self.world // pp // vtp, 1)
            other_mem[vtp] = states + act
        m_o, c_o = self.other_profile.fct_linear
        other_time = {}
        for vtp in vtp_opts:
            # vocab rows are data-parallel over the stage's remaining ranks
            # regardless of vtp (dp_v = stage_ranks/vtp): per-rank samples
            # do NOT shrink with vtp — only memory and comm layout change
            per_rank_samples = (global_bsz / chunks) / max(self.world // pp, 1)
            t = (m_o * per_rank_samples + c_o) * (1 + hw.bct_fct_coe)
            if vtp > 1:
                # vocab-tp is not free: the embedding output is allreduced
                # (masked-sum) over the vtp group, the head input is
                # (all)gathered from / re-split to the decoder layout at two
                # boundaries, and the collectives mirror in backward.  Price
                # fwd+bwd as 2 x (1 allreduce + 2 allgather-class moves) of
                # the per-dp-replica boundary activation.
                from .costmodel import _msg_latency
                dp_v = max((self.world // pp) // vtp, 1)
                msg_mb = ((global_bsz / chunks / dp_v) * lp.seq_length *
                          lp.hidden_size *
                          (2 if self.mixed_precision else 4) / (1024 * 1024))
                ar = hw.allreduce_latency_per_mb.get(f"{vtp}_1", 0.0)
                sel = hw.allgather_latency.get(vtp)
                ag = _msg_latency(sel, msg_mb) if sel else msg_mb * ar / 2
                t += 2 * (msg_mb * ar + 2 * ag)
            other_time[vtp] = t

        # per-layer-TYPE intra cost (with and without grad sync) per
        # strategy (reference: multi-layer-type DP)
        intra_sync: List[np.ndarray] = []
        intra_nosync: List[np.ndarray] = []
        mem_per_strategy: List[Dict[int, np.ndarray]] = []
        inter: List[np.ndarray] = []
        for tlp in profiles:
            isyn = np.zeros(S)
            inos = np.zeros(S)
            for si, s in enumerate(strategies):
                isyn[si] = layer_time_cost(
                    s, tlp, hw, global_bsz, chunks, self.world,
                    self.mixed_precision, no_gradient_sync=False)
                inos[si] = layer_time_cost(
                    s, tlp, hw, global_bsz, chunks, self.world,
                    self.mixed_precision, no_gradient_sync=True)
            intra_sync.append(isyn)
            intra_nosync.append(inos)
            mem_per_strategy.append({
                stage: np.array([
                    layer_memory_cost(
                        s, tlp, global_bsz, chunks, stage,
                        self.cfg.parallel.pipeline_type,
                        self.mixed_precision,
                        self.cfg.parallel.reduce_grads_each_microbatch,
                    )["total"]
                    for s in strategies])
                for stage in range(pp)})
            it = np.zeros((S, S))
            for a_i, sa in enumerate(strategies):
                for b_i, sb in enumerate(strategies):
                    mb = transition_cost_mb(sa, sb, tlp.seq_length,
                                            tlp.hidden_size,
                                            global_bsz / chunks / sb.dp,
                                            self.mixed_precision)
                    lat = hw.allreduce_latency_per_mb.get(
                        f"{max(sb.tp_sp * sb.cp, sa.tp_sp * sa.cp)}_1",
                        0.001)
                    it[a_i, b_i] = mb * lat
            inter.append(it)

        best: Optional[SearchResult] = None
        for division in self._division_candidates(pp):
          for vtp in vtp_opts:
            budget_units = int((budget_mb - other_mem[vtp]) // self.mem_unit)
            if budget_units <= 0:
                continue
            stage_paths: List[List[int]] = []
            stage_nosync: List[float] = []
            stage_sync: List[float] = []
            feasible = True
            lo = 0
            for stage in range(pp):
                n_lay = division[stage]
                types = [ltypes[lo + l] for l in range(n_lay)]
                lo += n_lay
                v_data = np.stack([
                    np.ceil(mem_per_strategy[t][stage] /
                            self.mem_unit).astype(np.int32) for t in types])
                intra_t = np.stack([intra_nosync[t] for t in types])
                inter_t = np.stack([inter[t] for t in types])
                inter_t[0] = 0.0
                f, mark = solve_layer_dp(v_data, intra_t, inter_t,
                                         budget_units)
                cost, path, _ = backtrace(v_data, mark, f, budget_units)
                if path is None:
                    feasible = False
                    break
                stage_paths.append(path)
                stage_nosync.append(cost)
                stage_sync.append(cost + float(sum(
                    intra_sync[t][p] - intra_nosync[t][p]
                    for t, p in zip(types, path))))
            if not feasible:
                continue
            # p2p term per stage boundary
            p2p = [layer_p2p_cost(strategies[stage_paths[i][-1]], lp, hw,
                                  global_bsz, chunks, self.mixed_precision)
                   for i in range(pp)]
            other_t = [other_time[vtp] if i in (0, pp - 1) else 0.0
                       for i in range(pp)]
            t_ms = pipeline_cost(
                [a + b for a, b in zip(stage_nosync, p2p)], stage_sync,
                chunks, pp, other_t)
            thr = global_bsz / (t_ms / 1000.0)
            if best is None or thr > best.throughput:
                flat = [strategies[p] for path in stage_paths for p in path]
                plan = self._plan_from(flat, pp, division, global_bsz,
                                       chunks, vtp)
                best = SearchResult(thr, plan, t_ms, global_bsz, chunks, pp,
                                    vtp)
        return best

    def _division_candidates(self, pp: int) -> List[List[int]]:
        """Even split + memory-balanced variants that relieve the edge
        stages (stage 0 carries the embedding's states/activations, the
        last stage the lm head) by shifting one layer inward
        (reference: memory-balanced pp division, search_engine.py:954)."""
        even = self._even_division(pp)
        if pp <= 1 or self.num_layers <= pp:
            return [even]
        cands = [even]

        def shift(frm: int, to: int, base: List[int]) -> None:
            d = list(base)
            if d[frm] > 1:
                d[frm] -= 1
                d[to] += 1
                if d not in cands:
                    cands.append(d)

        shift(0, 1, even)
        shift(pp - 1, pp - 2, even)
        if pp > 2:
            d2 = list(even)
            if d2[0] > 1 and d2[-1] > 1:
                d2[0] -= 1
                d2[-1] -= 1
                d2[1] += 1
                d2[-2] += 1
                if d2 not in cands:
                    cands.append(d2)
        return cands

    def _even_division(self, pp: int) -> List[int]:
        base = self.num_layers // pp
        rem = self.num_layers - base * pp
        return [base + (1 if i >= pp - rem else 0) for i in range(pp)]

    def _plan_from(self, flat: List[LayerStrategy], pp: int,
                   division: List[int], global_bsz: int, chunks: int,
                   vtp: int) -> HybridParallelPlan:
        return HybridParallelPlan(
            pp_deg=pp,
            tp_sizes_enc=[s.tp_sp for s in flat],
            tp_consecutive_flags=[1] * len(flat),
            cp_sizes_enc=[s.cp for s in flat],
            dp_types_enc=[1 if s.dp_type == "zero3" else 0 for s in flat],
            use_sp=[1 if s.use_ulysses else 0 for s in flat],
            checkpoint_flags=[1 if s.checkpoint else 0 for s in flat],
            pp_division=division,
            global_bsz=global_bsz, chunks=chunks,
            pipeline_type=self.cfg.parallel.pipeline_type,
            default_dp_type=self.cfg.parallel.default_dp_type,
            vtp=vtp, vsp=0, vcp=1)

    def check_cost_model(self, global_bsz: Optional[int] = None,
                         chunks: int = 1, pp: int = 1) -> str:
        """Cost-model introspection (reference search_engine.py:788
        check_cost_model): per-strategy time/memory for each layer type
        at one task point, as a printable table — the debugging view for
        why the DP picked what it picked."""
        gbsz = global_bsz or (self.args.settle_bsz
                              if self.args.settle_bsz > 0 else 8)
        strategies = enumerate_strategies(
            self.world, self.args, pp,
            default_dp_type=self.cfg.parallel.default_dp_type)
        lines = [f"cost model @ gbsz={gbsz} chunks={chunks} pp={pp} "
                 f"world={self.world} (budget {self.args.memory_constraint} GB)",
                 f"{'strategy':<34}{'t_sync ms':>10}{'t_nosync':>10}"
                 f"{'states MB':>11}{'act MB':>9}{'total MB':>10}"]
        for t, lp in enumerate(self.layer_profiles):
            if len(self.layer_profiles) > 1:
                lines.append(f"-- layer type {t}")
            for st in strategies:
                tsync = layer_time_cost(st, lp, self.hw, gbsz, chunks,
                                        self.world, self.mixed_precision,
                                        no_gradient_sync=False)
                tnos = layer_time_cost(st, lp, self.hw, gbsz, chunks,
                                       self.world, self.mixed_precision,
                                       no_gradient_sync=True)
                mem = layer_memory_cost(
                    st, lp, gbsz, chunks, 0,
                    self.cfg.parallel.pipeline_type, self.mixed_precision,
                    self.cfg.parallel.reduce_grads_each_microbatch)
                name = (f"tp{st.tp}sp{st.sp}cp{st.cp}dp{st.dp}"
                        f"-{st.dp_type}{'-ckpt' if st.checkpoint else ''}")
                lines.append(f"{name:<34}{tsync:>10.3f}{tnos:>10.3f}"
                             f"{mem['model_states']:>11.0f}"
                             f"{mem['activation']:>9.0f}"
                             f"{mem['total']:>10.0f}")
        return "\n".join(lines)

    def parallelism_optimization(self, output_path: Optional[str] = None
                                 ) -> Optional[SearchResult]:
        """Full task-grid search; saves the best plan JSON."""
        assert self.layer_profile is not None and self.hw is not None, \
            "load_profiles (or pass profiles) before searching"
        best: Optional[SearchResult] = None
        tasks = [(bsz, chunks, pp)
                 for bsz in self._bsz_candidates()
                 for pp in self._pp_candidates()
                 for chunks in self._chunk_candidates(bsz, pp)]
        if getattr(self.args, "parallel_search", False) and len(tasks) > 1:
            # the C++ DP core drops the GIL, so a thread pool overlaps the
            # per-task O(L*M*S^2) solves (reference: thread-parallel task
            # grid, search_engine.py:579-604)
            from concurrent.futures import ThreadPoolExecutor
            with ThreadPoolExecutor(max_workers=min(8, len(tasks))) as exe:
                results = list(exe.map(lambda t: self.search_task(*t),
                                       tasks))
        else:
            results = [self.search_task(*t) for t in tasks]
        for r in results:
            if r is not None:
                self.results.append(r)
                if best is None or r.throughput > best.throughput:
                    best = r
        if best is not None:
            path = output_path or self.args.output_config_path
            if path:
                cfgd = best.plan.to_config_dict()
                cfgd["searched_throughput_samples_per_sec"] = best.throughput
                write_json_config(cfgd, path)
        return best
