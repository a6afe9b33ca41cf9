#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract).

`python bench.py --gpus N --steps K --warmup W` runs a Llama-3-8B training
step (synthetic data, random-init weights, bf16) on N GPUs of one node.
For N>1 the driver launches it under torch.distributed.run, one rank per
GPU over RCCL; ranks read RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env.

Rank 0 prints exactly ONE JSON line with the whole-job tokens/sec
(BASELINE.json metric: "tokens/sec (whole node), auto-searched plan,
Llama-3-8B on 1/2/4/8 MI355X"). Weak scaling: per-GPU work fixed as N
grows (global_batch = per-gpu-batch * N).
"""
import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist


PROFILE_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "profiles")


def bench_layer_profiles(cfg, args):
    """LayerProfile/OtherProfile from the MEASURED model profiles committed
    under profiles/ (cli/profile_model output on MI355X); analytic
    fallback only if the files are absent (fresh checkout)."""
    from hetu_galvatron_amd.search.costmodel import LayerProfile, OtherProfile
    from hetu_galvatron_amd.search.engine import read_computation_profile

    m = cfg.model
    comp_path = os.path.join(
        PROFILE_DIR, f"computation_profiling_bf16_{args.model}.json")
    mem_path = os.path.join(
        PROFILE_DIR, f"model_profile_bf16_{args.model}.json")

    other_fct = None
    if os.path.exists(comp_path):
        from hetu_galvatron_amd.search.engine import (
            read_other_computation_profile)
        fct = read_computation_profile(comp_path, m.seq_length)
        other_fct = read_other_computation_profile(comp_path, m.seq_length)
    else:
        # analytic fallback: ~7.3 ms fwd per decoder layer per local sample
        # at seq 4096 (rocprofv3, profiles/bench_r01_kernel_stats_v2.md)
        print(f"[bench] {comp_path} missing; analytic fct fallback",
              file=sys.stderr)
        fct = (7.3 / 4096 * m.seq_length, 0.05)

    if os.path.exists(mem_path):
        with open(mem_path) as f:
            mem = json.load(f)
        lt = mem["layertype_0"]
        other = mem.get("other", {})
        lp = LayerProfile(
            parameter_mb=float(lt["parameter_size"]),
            fct_linear=fct,
            act_per_bsz_mb=lt["tp_activation_per_bsz_dict"],
            seq_length=m.seq_length, hidden_size=m.hidden_size)
        op = OtherProfile(
            parameter_mb=float(other.get("parameter_size", 0.0)),
            act_per_bsz_mb=other.get("tp_activation_per_bsz_dict", {"1": 0.0}),
            fct_linear=tuple(other.get("fct_linear",
                                       other_fct or (0.6, 0.05))))
    else:
        print(f"[bench] {mem_path} missing; analytic memory fallback",
              file=sys.stderr)
        lp = LayerProfile(
            parameter_mb=(4 * m.hidden_size ** 2 +
                          3 * m.hidden_size * m.ffn_hidden_size) * 4 / 1e6,
            fct_linear=fct,
            act_per_bsz_mb={str(t): m.seq_length * m.hidden_size * 2 * 18
                            / 1e6 / t for t in (1, 2, 4, 8)}
            | {"checkpoint": m.seq_length * m.hidden_size * 2 / 1e6},
            seq_length=m.seq_length, hidden_size=m.hidden_size)
        op = OtherProfile(
            parameter_mb=m.vocab_size * m.hidden_size * 4 / 1e6,
            act_per_bsz_mb={"1": m.seq_length * m.vocab_size * 2 / 1e6
                            / max(cfg.parallel.chunks, 1)},
            fct_linear=(0.6, 0.05))
    return lp, op


def search_bench_plan(cfg, world, args):
    """Run the FULL strategy search for the bench config; returns a plan.

    The space is unclamped: per-layer TP (Megatron-SP) / Ulysses-SP /
    DP / ZeRO-3 / activation-ckpt, pp in {1..world}, vocab-tp variants —
    the same task grid the reference searches (search_engine.py:520-575).
    Compute/memory inputs are measured profiles (profiles/); the xGMI
    collective model is measured hardware_configs when present, analytic
    otherwise (1-GPU leases cannot measure 8-GPU collectives).
    """
    from hetu_galvatron_amd.cli.search import default_mi355x_hardware
    from hetu_galvatron_amd.search.engine import (SearchEngine,
                                                  read_hardware_profiles)

    lp, op = bench_layer_profiles(cfg, args)
    world = max(world, 1)
    hw_dir = os.path.join(PROFILE_DIR, "hardware_configs")
    hw_file = os.path.join(
        hw_dir, f"allreduce_bandwidth_1nodes_{world}gpus_per_node.json")
    if os.path.exists(hw_file):
        hw = read_hardware_profiles(hw_dir, 1, world)
    else:
        hw = default_mi355x_hardware(world)
    cfg.search.num_nodes = 1
    cfg.search.num_gpus_per_node = world
    # 288 GB physical; 250 leaves headroom for allocator fragmentation and
    # CE/logit transients the memory model carries only approximately —
    # the plans this selects measured 212 GiB peak at N=1
    cfg.search.memory_constraint = 250
    cfg.search.settle_bsz = cfg.train.global_train_batch_size
    if args.chunks:
        cfg.search.settle_chunks = args.chunks
    eng = SearchEngine(cfg, lp, op, hw)
    out_path = None
    if os.path.isdir("gpurun_out") and int(os.environ.get("RANK", "0")) == 0:
        out_path = f"gpurun_out/bench_searched_plan_n{world}.json"  # provenance
    best = eng.parallelism_optimization(out_path)
    if best is None:
        return None
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] searched plan: pp{best.plan.pp_deg} "
              f"tp{best.plan.tp_sizes_enc[:4]}... "
              f"sp{best.plan.use_sp[:4]}... "
              f"zero3-flags{best.plan.dp_types_enc[:4]}... "
              f"ckpt{best.plan.checkpoint_flags[:4]}... "
              f"chunks={best.plan.chunks} vtp={best.plan.vtp} "
              f"predicted {best.throughput:.2f} samples/s", file=sys.stderr)
    return best.plan


def plan_desc(plan, world: int) -> str:
    """Compact human-readable summary of the executed per-layer plan."""
    n = plan.num_layers
    tps = sorted(set(plan.tp_sizes_enc))
    tp_s = str(tps[0]) if len(tps) == 1 else f"{tps[0]}-{tps[-1]}"
    per_stage = world // plan.pp_deg
    dps = sorted({per_stage // (t * c) for t, c in
                  zip(plan.tp_sizes_enc, plan.cp_sizes_enc)})
    dp_s = str(dps[0]) if len(dps) == 1 else f"{dps[0]}-{dps[-1]}"
    return (f"searched:pp{plan.pp_deg}-tp{tp_s}-dp{dp_s}-"
            f"{plan.default_dp_type}"
            f"-sp{sum(plan.use_sp)}/{n}-zero3:{sum(plan.dp_types_enc)}/{n}"
            f"-ckpt:{sum(plan.checkpoint_flags)}/{n}-chunks{plan.chunks}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--model", type=str, default="llama-3-8b")
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--batch-per-gpu", type=int, default=8,
                    help="sequences per GPU per step (weak scaling)")
    ap.add_argument("--chunks", type=int, default=0,
                    help="pipeline microbatches (0 = auto: max(2, 2*pp))")
    ap.add_argument("--pp", type=int, default=1)
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--dp-type", type=str, default="zero2",
                    choices=["ddp", "zero2", "zero3"])
    ap.add_argument("--checkpoint", action="store_true")
    ap.add_argument("--plan", type=str, default="auto",
                    help="searched-plan JSON path, 'auto' (run the search "
                         "engine), or 'none' (uniform knobs)")
    ap.add_argument("--tune-gemms", type=str, default=None,
                    help="TunableOp tuning during warmup; CSV written here "
                         "at exit (then commit as profiles/tunableop_gfx950.csv)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(world, args.gpus) if world > 1 else args.gpus

    use_gpu = torch.cuda.is_available()
    if not use_gpu:
        print("bench.py requires a GPU", file=sys.stderr)
        sys.exit(1)
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    # hipBLASLt/rocBLAS algorithm selection tuned offline on MI355X for the
    # bench GEMM shapes (tools/tune_gemms.py -> profiles/tunableop_gfx950.csv):
    # TunableOp in read-only mode replays the tuned picks, no runtime sweep
    tuned = os.path.join(PROFILE_DIR, "tunableop_gfx950.csv")
    if args.tune_gemms:
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(True)
        torch.cuda.tunable.set_filename(args.tune_gemms,
                                        insert_device_ordinal=False)
        torch.cuda.tunable.set_max_tuning_duration(10)
        torch.cuda.tunable.set_max_tuning_iterations(10)
        if os.path.exists(tuned):
            torch.cuda.tunable.read_file(tuned)  # warm-start from committed
    elif os.path.exists(tuned) and os.environ.get("GALVATRON_NO_TUNABLEOP") != "1":
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(False)
        torch.cuda.tunable.set_filename(tuned, insert_device_ordinal=False)
        if not torch.cuda.tunable.read_file(tuned):
            print(f"[bench] tunableop read_file failed: {tuned}",
                  file=sys.stderr)

    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)
    from hetu_galvatron_amd.runtime.galvatron_model import resolve_plan

    global_batch = args.batch_per_gpu * max(world, 1)
    chunks = args.chunks or max(2, 2 * args.pp)
    cfg = load_config(base={
        "model": {"model_name": args.model, "seq_length": args.seq_len},
        "train": {"global_train_batch_size": global_batch,
                  "train_iters": args.steps + args.warmup,
                  "lr": 1e-4, "lr_decay_style": "constant",
                  "distributed_backend": "nccl"},
        "parallel": {"pp_deg": args.pp, "global_tp_deg": args.tp,
                     "default_dp_type": args.dp_type,
                     "global_checkpoint": 1 if args.checkpoint else 0,
                     "chunks": chunks, "mixed_precision": "bf16",
                     "galvatron_config_path":
                         args.plan if args.plan not in ("auto", "none", "")
                         else None},
    })

    if world > 1:
        initialize_galvatron(cfg, backend="nccl")
    else:
        torch.manual_seed(cfg.train.seed)

    if args.plan == "auto" and args.pp == 1 and args.tp == 1:
        # auto-searched plan (BASELINE metric): full per-layer
        # TP/SP/DP/ZeRO-3/ckpt x pp x vtp space under the 288 GB budget,
        # measured per-layer compute/memory profiles (profiles/)
        plan = search_bench_plan(cfg, world, args)
        if plan is None:
            plan = resolve_plan(cfg, world)
    else:
        plan = resolve_plan(cfg, world)
    model = GalvatronModel(cfg, plan, device=device)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(cfg, device)

    def one_step():
        opt.zero_grad()
        ctx = next(it)
        stats = model.forward_backward(ctx)
        opt.step()
        sched.step()
        return stats

    for _ in range(args.warmup):
        one_step()
    if args.tune_gemms:
        # freeze the tuned selections: the timed region below measures the
        # replayed picks, never an in-flight sweep
        torch.cuda.tunable.tuning_enable(False)

    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks = whole-job wall time
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = args.steps * global_batch * args.seq_len
    value = tokens / elapsed
    if rank == 0 and torch.cuda.is_available():
        peak = torch.cuda.max_memory_allocated() / (1 << 30)
        print(f"[bench] peak GPU memory: {peak:.1f} GiB", file=sys.stderr)
    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec (whole node), auto-searched plan, "
                      "Llama-3-8B on 1/2/4/8 MI355X",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": max(world, 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.seq_len,
                "parallelism": plan_desc(plan, max(world, 1)),
            },
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
