"""Strategy-search entry (reference: galvatron/models/gpt/search_dist.py:11).

  python -m hetu_galvatron_amd.cli.search [cfg.yaml] \\
      model.model_name=llama-3-8b search.memory_constraint=240 \\
      search.output_config_path=configs/galvatron_config_llama3-8b.json

Profiled inputs: profile.profile_dir for the model profile
(model_profile_{prec}_{model}.json + computation_profiling_...) and
profile_hardware.hardware_config_dir for the bandwidth JSONs. Falls back
to an analytic MI355X default hardware profile when no measured JSONs
exist (measure with cli.profile_hardware for accurate plans).
"""
from __future__ import annotations

import json
import os
import sys


def default_mi355x_hardware(world: int):
    """Analytic xGMI defaults (7 links x ~153 GB/s per GPU) used until a
    measured profile exists."""
    from ..profiler.hardware_bench import _groups_for  # noqa: F401
    from ..search.costmodel import HardwareProfile
    hw = HardwareProfile()
    for n in (2, 4, 8):
        bw = 300.0 if n > 2 else 150.0  # bus GB/s estimates
        for c in (0, 1):
            hw.allreduce_latency_per_mb[f"{n}_{c}"] = \
                2 * (n - 1) / n / (bw * 1.024)
        hw.allgather_latency[n] = {"popt": (1.0 / (bw * 1.024) / 2, 0.03)}
        hw.all2all_latency[n] = {"popt": (1.0 / (bw * 1.024) / 2, 0.03)}
    for p in (2, 4, 8):
        hw.p2p_latency_per_mb[p] = 1.0 / (140.0 * 1.024)
    return hw


def main(argv=None):
    from ..config.loader import config_from_cli
    from ..search.costmodel import LayerProfile, OtherProfile
    from ..search.engine import (SearchEngine, read_computation_profile,
                                 read_hardware_profiles)

    argv = list(argv) if argv is not None else sys.argv[1:]
    # cost-model introspection mode (reference search_engine.py:788
    # check_cost_model): print per-strategy time/memory instead of searching
    check_cm = "--check-cost-model" in argv
    if check_cm:
        argv.remove("--check-cost-model")
    cfg = config_from_cli(argv)
    prec = "bf16" if cfg.parallel.mixed_precision == "bf16" else "fp32"
    name = cfg.model.model_name or "model"
    pdir = cfg.profile.profile_dir
    comp_path = os.path.join(pdir, f"computation_profiling_{prec}_{name}.json")
    mem_path = os.path.join(pdir, f"model_profile_{prec}_{name}.json")
    hw_dir = cfg.profile_hardware.hardware_config_dir

    lp = op = hw = None
    if os.path.exists(mem_path):
        with open(mem_path) as f:
            mem = json.load(f)
        fct = (read_computation_profile(comp_path, cfg.model.seq_length)
               if os.path.exists(comp_path) else (1.0, 0.1))
        lt = mem["layertype_0"]
        lp = LayerProfile(parameter_mb=float(lt["parameter_size"]),
                          fct_linear=fct,
                          act_per_bsz_mb=lt["tp_activation_per_bsz_dict"],
                          seq_length=cfg.model.seq_length,
                          hidden_size=cfg.model.hidden_size)
        other = mem.get("other", {})
        op = OtherProfile(
            parameter_mb=float(other.get("parameter_size", 0.0)),
            act_per_bsz_mb=other.get("tp_activation_per_bsz_dict", {}),
            fct_linear=tuple(other.get("fct_linear", (0.0, 0.0))))
    else:
        print(f"[search] no model profile at {mem_path}; "
              "using analytic estimates — run cli.profile_model first "
              "for accurate plans", file=sys.stderr)
        m = cfg.model
        param_mb = (4 * m.hidden_size * m.hidden_size +
                    3 * m.hidden_size * m.ffn_hidden_size) * 4 / 1e6
        act1 = m.seq_length * m.hidden_size * 2 * 18 / 1e6
        lp = LayerProfile(
            parameter_mb=param_mb, fct_linear=(param_mb / 1000.0, 0.2),
            act_per_bsz_mb={str(t): act1 / t for t in (1, 2, 4, 8)}
            | {"checkpoint": m.seq_length * m.hidden_size * 2 / 1e6},
            seq_length=m.seq_length, hidden_size=m.hidden_size)
        op = OtherProfile(parameter_mb=m.vocab_size * m.hidden_size * 4 / 1e6,
                          act_per_bsz_mb={"1": act1},
                          fct_linear=(0.3, 0.05))

    world = cfg.search.num_nodes * cfg.search.num_gpus_per_node
    if os.path.exists(os.path.join(
            hw_dir, f"allreduce_bandwidth_{cfg.search.num_nodes}nodes_"
            f"{cfg.search.num_gpus_per_node}gpus_per_node.json")):
        hw = read_hardware_profiles(hw_dir, cfg.search.num_nodes,
                                    cfg.search.num_gpus_per_node)
    else:
        print(f"[search] no hardware profile in {hw_dir}; using analytic "
              "xGMI defaults — run cli.profile_hardware first",
              file=sys.stderr)
        hw = default_mi355x_hardware(world)

    eng = SearchEngine(cfg, lp, op, hw)
    if check_cm:
        print(eng.check_cost_model())
        return
    out = cfg.search.output_config_path or os.path.join(
        pdir, f"galvatron_config_{name}_{cfg.search.num_nodes}nodes_"
        f"{cfg.search.num_gpus_per_node}gpus_per_node_"
        f"{cfg.search.memory_constraint}GB_{prec}.json")
    best = eng.parallelism_optimization(out)
    if best is None:
        print("[search] no feasible plan under the memory budget",
              file=sys.stderr)
        sys.exit(1)
    print(json.dumps({
        "throughput_samples_per_sec": best.throughput,
        "time_ms": best.time_ms, "pp_deg": best.pp_deg,
        "chunks": best.chunks, "global_bsz": best.global_bsz,
        "vtp": best.vtp, "plan_path": out}, indent=2))


if __name__ == "__main__":
    main()
