"""torchrun worker measuring RCCL-over-xGMI collective performance.

Reference behavior: galvatron/profile_hardware/profile_allreduce.py:20-270,
profile_p2p.py:19-225, profile_all2all.py, profile_overlap.py:10-192 —
one worker, subcommand per sweep.  Writes the hardware_configs JSONs the
search engine consumes (same keys: "allreduce_size_{n}_consec_{c}",
"pp_size_{n}", "allreduce_size_{n}_{m}MB_time", "overlap_coe").

Launch (per sweep):
  python -m torch.distributed.run --nnodes 1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 -m hetu_galvatron_amd.profiler.hardware_bench \
      --op allreduce --output-dir hardware_configs
"""
from __future__ import annotations

import argparse
import json
import os
import time
from typing import Dict, List

import torch
import torch.distributed as dist


def _init():
    rank = int(os.environ.get("RANK", "0"))
    local = int(os.environ.get("LOCAL_RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(local)
    dist.init_process_group(backend)
    return rank, local, world, backend


def _dev():
    return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")


def _sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def _time_op(fn, warmup: int, iters: int) -> float:
    for _ in range(warmup):
        fn()
    _sync()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    _sync()
    return (time.perf_counter() - t0) / iters


def _groups_for(world: int, size: int, consec: bool) -> List[List[int]]:
    """Rank lists for sub-groups of `size` (consecutive or strided).
    Matches the reference's consec 1/0 axis = xGMI-hop asymmetry on MI355X."""
    if consec:
        return [list(range(i, i + size)) for i in range(0, world, size)]
    stride = world // size
    return [list(range(i, world, stride)) for i in range(stride)]


def bench_allreduce(args, rank, world) -> Dict[str, float]:
    out = {}
    dev = _dev()
    dtype = torch.bfloat16
    numel = args.message_mb * 1024 * 1024 // 2
    buf = torch.randn(numel, dtype=dtype, device=dev)
    for size in [s for s in (world, world // 2, world // 4, 2) if 2 <= s <= world]:
        for consec in ([True] if size == world else [True, False]):
            groups = _groups_for(world, size, consec)
            my = None
            for ranks in groups:
                g = dist.new_group(ranks)
                if rank in ranks:
                    my = g
            dist.barrier()
            if my is not None:
                t = _time_op(lambda: dist.all_reduce(buf, group=my),
                             args.warmup_iters, args.measure_iters)
            else:
                t = 0.0
            tt = torch.tensor([t], dtype=torch.float64,
                              device=dev if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(tt, op=dist.ReduceOp.MAX)
            t = float(tt.item())
            bus_gb = (2 * (size - 1) / size) * args.message_mb / 1024 / t \
                if t > 0 else 0.0
            key = f"allreduce_size_{size}_consec_{1 if consec else 0}"
            out[key] = round(bus_gb, 3)
            if size == world:
                out[f"allreduce_size_{size}_consec_0"] = out[key]
    return out


def bench_p2p(args, rank, world) -> Dict[str, float]:
    out = {}
    dev = _dev()
    numel = args.message_mb * 1024 * 1024 // 2
    buf = torch.randn(numel, dtype=torch.bfloat16, device=dev)
    for pp in [p for p in (2, 4, 8) if p <= world]:
        per = world // pp
        # even stages send to the next stage; odd stages receive
        send = (rank // per) % 2 == 0 and rank + per < world

        def step():
            ops = []
            if send and rank + per < world:
                ops.append(dist.P2POp(dist.isend, buf, rank + per))
            elif not send and rank - per >= 0:
                ops.append(dist.P2POp(dist.irecv, buf, rank - per))
            if ops:
                for w in dist.batch_isend_irecv(ops):
                    w.wait()
        dist.barrier()
        t = _time_op(step, args.warmup_iters, args.measure_iters)
        tt = torch.tensor([t], dtype=torch.float64,
                          device=dev if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        t = float(tt.item())
        out[f"pp_size_{pp}"] = round(args.message_mb / 1024 / t, 3) if t > 0 else 0.0
    return out


def bench_sp_time(args, rank, world) -> Dict[str, float]:
    """allreduce + all2all latency vs message size, PER SUB-GROUP SIZE
    (the cost model needs latency curves for every tp/sp degree —
    reference sp_time_* sweeps sizes, not just the full world)."""
    out = {}
    dev = _dev()
    sizes = [s for s in (2, 4, 8, 16) if s <= world] or [world]
    if world not in sizes and world > 1:
        sizes.append(world)
    for size in sizes:
        groups = _groups_for(world, size, True)
        my = None
        for ranks in groups:
            g = dist.new_group(ranks)
            if rank in ranks:
                my = g
        dist.barrier()
        mb = args.start_mb
        while mb <= args.end_mb:
            numel = mb * 1024 * 1024 // 2
            buf = torch.randn(numel, dtype=torch.bfloat16, device=dev)
            obuf = torch.empty_like(buf)
            t_ar = _time_op(lambda: dist.all_reduce(buf, group=my),
                            args.warmup_iters, args.measure_iters) \
                if my is not None else 0.0
            try:
                t_a2a = _time_op(
                    lambda: dist.all_to_all_single(obuf, buf, group=my),
                    args.warmup_iters, args.measure_iters) \
                    if my is not None else 0.0
            except RuntimeError:
                t_a2a = 0.0  # gloo: no all_to_all_single
            tt = torch.tensor([t_ar, t_a2a], dtype=torch.float64,
                              device=dev if dist.get_backend() == "nccl"
                              else "cpu")
            dist.all_reduce(tt, op=dist.ReduceOp.MAX)
            out[f"allreduce_size_{size}_{mb}MB_time"] = \
                round(float(tt[0]) * 1000, 5)
            if float(tt[1]) > 0:
                out[f"all2all_size_{size}_{mb}MB_time"] = \
                    round(float(tt[1]) * 1000, 5)
            mb *= 2
    return out


def bench_overlap(args, rank, world) -> Dict[str, float]:
    """Comm/compute overlap slowdown coefficient
    (reference profile_overlap.py: concurrent streams vs isolated)."""
    dev = _dev()
    if not torch.cuda.is_available():
        return {"overlap_coe": 1.15}
    n = 2048
    a = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
    b = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
    buf = torch.randn(64 * 1024 * 1024 // 2, dtype=torch.bfloat16, device=dev)

    t_comp = _time_op(lambda: torch.mm(a, b), 5, 20)
    comm_stream = torch.cuda.Stream()

    def overlapped():
        with torch.cuda.stream(comm_stream):
            dist.all_reduce(buf)
        for _ in range(4):
            torch.mm(a, b)
        torch.cuda.current_stream().wait_stream(comm_stream)

    t_both = _time_op(overlapped, 5, 20)
    coe = max(t_both / (4 * t_comp), 1.0)
    tt = torch.tensor([coe], dtype=torch.float64, device=dev)
    dist.all_reduce(tt)
    return {"overlap_coe": float(tt.item()) / world}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", required=True,
                    choices=["allreduce", "p2p", "sp_time", "overlap"])
    ap.add_argument("--output-dir", default="hardware_configs")
    ap.add_argument("--message-mb", type=int, default=512)
    ap.add_argument("--start-mb", type=int, default=1)
    ap.add_argument("--end-mb", type=int, default=1024)
    ap.add_argument("--warmup-iters", type=int, default=5)
    ap.add_argument("--measure-iters", type=int, default=20)
    ap.add_argument("--num-nodes", type=int, default=1)
    args = ap.parse_args()

    rank, local, world, backend = _init()
    fn = {"allreduce": bench_allreduce, "p2p": bench_p2p,
          "sp_time": bench_sp_time, "overlap": bench_overlap}[args.op]
    result = fn(args, rank, world)
    if rank == 0:
        os.makedirs(args.output_dir, exist_ok=True)
        gpus = world // args.num_nodes
        name = {"allreduce": "allreduce_bandwidth", "p2p": "p2p_bandwidth",
                "sp_time": "sp_time", "overlap": "overlap_coefficient"}[args.op]
        if args.op == "overlap":
            path = os.path.join(args.output_dir, "overlap_coefficient.json")
        else:
            path = os.path.join(
                args.output_dir,
                f"{name}_{args.num_nodes}nodes_{gpus}gpus_per_node.json")
        merged = {}
        if os.path.exists(path):
            with open(path) as f:
                merged = json.load(f)
        merged.update(result)
        with open(path, "w") as f:
            json.dump(merged, f, indent=4)
        print(f"[hardware_bench] wrote {path}: {result}")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
