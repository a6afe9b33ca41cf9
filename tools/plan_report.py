"""Pretty-print a searched plan JSON (per-layer strategy table).

  python tools/plan_report.py examples/configs/galvatron_config_*.json \\
      [--world 8]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from hetu_galvatron_amd.config import HybridParallelPlan  # noqa: E402


def report(path: str, world: int) -> str:
    plan = HybridParallelPlan.load(path)
    lines = [f"{os.path.basename(path)}",
             f"  pp={plan.pp_deg} division={plan.pp_division} "
             f"gbsz={plan.global_bsz} chunks={plan.chunks} "
             f"{plan.pipeline_type} vtp={plan.vtp} vsp={plan.vsp} "
             f"vcp={plan.vcp}"]
    # group consecutive identical layers
    runs = []
    for i in range(plan.num_layers):
        s = plan.layer(i, world)
        key = (s.tp, s.sp, s.cp, s.dp, s.dp_type, s.checkpoint)
        if runs and runs[-1][0] == key:
            runs[-1][2] = i
        else:
            runs.append([key, i, i])
    for (tp, sp, cp, dp, dpt, ck), a, b in runs:
        rng = f"{a}" if a == b else f"{a}-{b}"
        mode = f"sp{sp}" if sp > 1 else f"tp{tp}"
        lines.append(f"  layers {rng:>7}: {mode} cp{cp} dp{dp} {dpt}"
                     f"{' +ckpt' if ck else ''}")
    return "\n".join(lines)


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("paths", nargs="+")
    ap.add_argument("--world", type=int, default=8)
    ns = ap.parse_args(argv)
    for p in ns.paths:
        print(report(p, ns.world))


if __name__ == "__main__":
    main()
