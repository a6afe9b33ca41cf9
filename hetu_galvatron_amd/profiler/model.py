"""Model profiler: memory/computation sweeps + post-processing.

Reference: galvatron/core/profiler/model_profiler.py:215-805
(launch_profiling_scripts spawning the training entry with overrides;
_process_computation_data: per-layer time = Δtime/Δlayers/bsz by layernum
differencing; _process_memory_data: parameter/activation separation by
layernum+tp regression).

Outputs (search-engine inputs):
  computation_profiling_{prec}_{model}.json   raw layernum[N]_bsz_seq keys
  memory_profiling_{prec}_{model}.json        raw layout/... keys
  model_profile_{prec}_{model}.json           parsed: layertype_0 + other
"""
from __future__ import annotations

import json
import os
import re
import subprocess
import sys
from typing import Dict, List, Optional, Tuple

from ..config import GalvatronConfig


class ModelProfiler:
    def __init__(self, cfg: GalvatronConfig):
        self.cfg = cfg
        self.p = cfg.profile
        self.prec = "bf16" if cfg.parallel.mixed_precision == "bf16" else "fp32"
        self.name = cfg.model.model_name or "model"
        self.dir = self.p.profile_dir

    # -- launching ---------------------------------------------------------
    def _run(self, overrides: List[str], nproc: int = 1) -> None:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes", "1",
               "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
               "--master-port", "29517",
               "-m", "hetu_galvatron_amd.cli.train"] + overrides
        env = dict(os.environ)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        subprocess.run(cmd, check=True, env=env)

    def _base_overrides(self) -> List[str]:
        return [f"model.model_name={self.name}",
                f"model.seq_length={self.cfg.model.seq_length}",
                "train.train_iters=5",
                "profile.profile=1",
                f"profile.profile_dir={self.dir}"]

    def launch_computation_profiling(self) -> None:
        """layernum x bsz sweep, 1 GPU (reference :343-420)."""
        p = self.p
        for layernum in (p.profile_min_layer_num, p.profile_max_layer_num):
            bsz = p.profile_batch_size_start
            while bsz <= p.profile_batch_size_end:
                # doubling sweep (reference uses halving 1024->1): the fct
                # fit is linear in bsz, so power-of-2 points suffice and
                # keep the GPU sweep short
                self._run(self._base_overrides() + [
                    "profile.profile_type=computation",
                    f"model.num_hidden_layers={layernum}",
                    f"train.global_train_batch_size={bsz}",
                    "parallel.chunks=1"])
                bsz *= 2

    def launch_memory_profiling(self, nproc: int = 8) -> None:
        """pp1 x tp x {ckpt} (+ pp sweeps) layouts (reference :231-343)."""
        p = self.p
        bsz = p.profile_fixed_batch_size
        for layernum in (p.profile_min_layer_num, p.profile_max_layer_num):
            tp = 1
            while tp <= min(nproc, 8):
                self._run(self._base_overrides() + [
                    "profile.profile_type=memory",
                    f"model.num_hidden_layers={layernum}",
                    f"train.global_train_batch_size={bsz}",
                    f"parallel.global_tp_deg={tp}", "parallel.chunks=1"],
                    nproc=nproc)
                tp *= 2
        # activation-checkpointing layout
        self._run(self._base_overrides() + [
            "profile.profile_type=memory",
            f"model.num_hidden_layers={p.profile_max_layer_num}",
            f"train.global_train_batch_size={bsz}",
            "parallel.global_checkpoint=1", "parallel.chunks=1"],
            nproc=nproc)

    # -- post-processing ---------------------------------------------------
    def process_computation_data(self, raw: Optional[Dict] = None
                                 ) -> Dict[str, float]:
        """Keeps the raw JSON as-is (the search engine's
        read_computation_profile does the layernum differencing)."""
        path = os.path.join(self.dir,
                            f"computation_profiling_{self.prec}_{self.name}.json")
        if raw is None:
            with open(path) as f:
                raw = json.load(f)
        return raw

    def process_memory_data(self, raw: Optional[Dict] = None,
                            write: bool = True) -> Dict:
        """Raw memory JSON -> parsed per-layer profile.

        Keys: "{pp}_{tp}_{dp}[_c]/layernum[N]_bsz{B}_seq{S}_rank{r}_{ms|act|act_peak}"
        parameter_size (fp32 MB): Δmodel-states / Δlayers / 4 at tp=1;
        activation per bsz per tp: Δact / Δlayers / bsz;
        other (embedding/head): layout totals minus per-layer extrapolation.
        """
        path = os.path.join(self.dir,
                            f"memory_profiling_{self.prec}_{self.name}.json")
        if raw is None:
            with open(path) as f:
                raw = json.load(f)

        pat = re.compile(
            r"(\d+)_(\d+)_(\d+)(_c)?/layernum\[(\d+)\]_bsz(\d+)_seq(\d+)"
            r"_rank(\d+)_(act_peak|act|ms)$")
        table: Dict[Tuple, float] = {}
        for k, v in raw.items():
            m = pat.match(k)
            if not m:
                continue
            pp, tp, dp, ck, ln, bsz, seq, rank, kind = m.groups()
            if rank != "0":
                continue
            table[(int(pp), int(tp), bool(ck), int(ln), int(bsz), kind)] = \
                float(v)

        lns = sorted({t[3] for t in table})
        assert len(lns) >= 2, f"need 2 layernum sweeps, got {lns}"
        n1, n2 = lns[0], lns[-1]
        dn = n2 - n1

        def get(pp, tp, ck, ln, kind, bsz=None):
            if bsz is None:
                c = [v for t, v in table.items()
                     if t[:4] == (pp, tp, ck, ln) and t[5] == kind]
                return c[0] if c else None
            return table.get((pp, tp, ck, ln, bsz, kind))

        # parameter size from tp=1 model-states differencing
        ms1, ms2 = get(1, 1, False, n1, "ms"), get(1, 1, False, n2, "ms")
        param_mb = (ms2 - ms1) / dn / 4.0
        other_param_mb = max((ms1 - n1 * 4.0 * param_mb) / 4.0, 0.0)

        bszs = sorted({t[4] for t in table})
        bsz = bszs[-1]
        act_dict: Dict[str, float] = {}
        tp = 1
        while True:
            a1 = get(1, tp, False, n1, "act")
            a2 = get(1, tp, False, n2, "act")
            if a1 is None or a2 is None:
                break
            act_dict[str(tp)] = max((a2 - a1) / dn / bsz, 1e-3)
            tp *= 2
        ac1 = get(1, 1, True, n1, "act")
        ac2 = get(1, 1, True, n2, "act")
        if ac1 is not None and ac2 is not None:
            act_dict["checkpoint"] = max((ac2 - ac1) / dn / bsz, 1e-3)
        else:
            act_dict.setdefault("checkpoint", act_dict.get("1", 1.0) * 0.07)

        a1 = get(1, 1, False, n1, "act")
        other_act = max((a1 - n1 * act_dict["1"] * bsz) / bsz, 0.0)

        parsed = {
            "layertype_0": {
                "parameter_size": param_mb,
                "tp_activation_per_bsz_dict": act_dict,
            },
            "other": {
                "parameter_size": other_param_mb,
                "tp_activation_per_bsz_dict": {"1": other_act},
            },
        }
        if write:
            out = os.path.join(self.dir,
                               f"model_profile_{self.prec}_{self.name}.json")
            os.makedirs(self.dir, exist_ok=True)
            with open(out, "w") as f:
                json.dump(parsed, f, indent=4)
        return parsed


class T5ModelProfiler(ModelProfiler):
    """Enc-dec (t5) profiling: two layer types via two-axis layernum
    differencing (reference: the multi-layer-type profiles consumed by
    _build_dp_and_run_multi_layer_type).

    Sweep points (enc, dec): (n1, n1), (n2, n1), (n1, n2) — the encoder
    delta comes from axis 0, the decoder delta from axis 1; raw keys are
    "layernum[enc,dec]_bsz{B}_seq{S}".  process_t5_data emits
    layertype_0 (encoder) / layertype_1 (decoder), each with its own
    fct_linear, ready for SearchEngine.load_profiles.
    """

    def _combos(self):
        n1, n2 = self.p.profile_min_layer_num, self.p.profile_max_layer_num
        return [(n1, n1), (n2, n1), (n1, n2)], n1, n2

    def launch_computation_profiling(self) -> None:
        p = self.p
        combos, _, _ = self._combos()
        for enc, dec in combos:
            for bsz in range(p.profile_batch_size_start,
                             p.profile_batch_size_end + 1):
                self._run(self._base_overrides() + [
                    "profile.profile_type=computation",
                    f"model.num_hidden_layers={enc}",
                    f"model.num_decoder_layers={dec}",
                    f"train.global_train_batch_size={bsz}",
                    "parallel.mixed_precision=" +
                    self.cfg.parallel.mixed_precision])

    def launch_memory_profiling(self, nproc: int = 8) -> None:
        combos, _, _ = self._combos()
        for enc, dec in combos:
            self._run(self._base_overrides() + [
                "profile.profile_type=memory",
                f"model.num_hidden_layers={enc}",
                f"model.num_decoder_layers={dec}",
                f"train.global_train_batch_size={self.p.profile_fixed_batch_size}",
            ], nproc=1)

    def process_t5_data(self, comp_raw: Dict, mem_raw: Dict,
                        write: bool = True) -> Dict:
        """Two-axis differencing -> layertype_0/1 profile JSON."""
        _, n1, n2 = self._combos()
        dn = n2 - n1
        seq = self.cfg.model.seq_length

        cpat = re.compile(r"layernum\[(\d+),(\d+)\]_bsz(\d+)_seq(\d+)$")
        by_pt: Dict[Tuple[int, int], Dict[int, float]] = {}
        for k, v in comp_raw.items():
            m = cpat.match(k)
            if not m or int(m.group(4)) != seq:
                continue
            by_pt.setdefault((int(m.group(1)), int(m.group(2))),
                             {})[int(m.group(3))] = float(v)
        base = by_pt[(n1, n1)]
        fcts = []
        for pt in ((n2, n1), (n1, n2)):
            xs, ys = [], []
            for b in sorted(set(base) & set(by_pt[pt])):
                xs.append(b)
                ys.append((by_pt[pt][b] - base[b]) / dn)
            m_, c_ = _fit(xs, ys)
            fcts.append((m_, c_))

        mpat = re.compile(
            r"(\d+)_(\d+)_(\d+)(_c)?/layernum\[(\d+),(\d+)\]_bsz(\d+)"
            r"_seq(\d+)_rank(\d+)_(act_peak|act|ms)$")
        table: Dict[Tuple, float] = {}
        for k, v in mem_raw.items():
            m = mpat.match(k)
            if not m or m.group(9) != "0":
                continue
            table[(int(m.group(5)), int(m.group(6)), m.group(10))] = float(v)
        bszs = {int(mpat.match(k).group(7)) for k in mem_raw
                if mpat.match(k)}
        bsz = max(bszs) if bszs else self.p.profile_fixed_batch_size

        out = {}
        for ti, pt in enumerate(((n2, n1), (n1, n2))):
            dms = (table[(pt[0], pt[1], "ms")] -
                   table[(n1, n1, "ms")]) / dn
            dact = (table[(pt[0], pt[1], "act")] -
                    table[(n1, n1, "act")]) / dn
            out[f"layertype_{ti}"] = {
                "parameter_size": max(dms / 4.0, 1e-3),
                "fct_linear": list(fcts[ti]),
                "tp_activation_per_bsz_dict": {
                    "1": max(dact / bsz, 1e-3),
                    "checkpoint": max(dact / bsz, 1e-3) * 0.07},
            }
        base_ms = table[(n1, n1, "ms")]
        per_layer_ms = (out["layertype_0"]["parameter_size"] +
                        out["layertype_1"]["parameter_size"]) * 4.0 * n1
        out["other"] = {
            "parameter_size": max((base_ms - per_layer_ms) / 4.0, 0.0),
            "tp_activation_per_bsz_dict": {"1": 1.0},
        }
        if write:
            path = os.path.join(
                self.dir, f"model_profile_{self.prec}_{self.name}.json")
            os.makedirs(self.dir, exist_ok=True)
            with open(path, "w") as f:
                json.dump(out, f, indent=4)
        return out


def _fit(xs, ys):
    import numpy as np
    m, c = np.polyfit(np.asarray(xs, float), np.asarray(ys, float), 1)
    return float(m), float(c)
