"""Distributed (per-rank sharded) checkpoint save / resume.

Reference: per-rank .pt files + hybrid_parallel_configs.json consistency
check (hybrid_parallel_config.py:132-144), optimizer per-rank state
(optimizer/utils.py:57-70), save_llama_module (llama_adapter.py:172).

Layout:
    <save>/latest_checkpointed_iteration.txt
    <save>/iter_0000010/hybrid_parallel_config.json
    <save>/iter_0000010/rank_00003.pt   # per-rank: masters + adam moments
                                        #   + scheduler + rng

Each rank saves its OWNED fp32 master shard + exp_avg/exp_avg_sq per
block; on load bf16 params regenerate from the masters (exact resume —
the bf16 copy is derived state). Resuming requires the SAME plan and
world size (checked against the stored plan JSON).
"""
from __future__ import annotations

import json
import os
from typing import Optional

import torch
import torch.distributed as dist


def _rank_world():
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def save_distributed_checkpoint(model, opt, sched, cfg, iteration: int,
                                save_dir: Optional[str] = None,
                                rerun_state_machine=None) -> str:
    save_dir = save_dir or cfg.ckpt.save
    rank, world = _rank_world()
    it_dir = os.path.join(save_dir, f"iter_{iteration:07d}")
    os.makedirs(it_dir, exist_ok=True)

    blocks = {}
    expert_blocks = {}
    for i, blk in enumerate(model.stage_model.blocks):
        # both flat-param partitions travel: dense (blk.flat) and, for MoE
        # layers under ep>1, the expert partition on the EDP group
        # (blk.flat_expert) — its fp32 masters + Adam moments are stepped by
        # optimizer.py:127 and must resume exactly like the dense ones
        for f, store in ((blk.flat, blocks),
                         (getattr(blk, "flat_expert", None), expert_blocks)):
            if f is None or f.total == 0:
                continue
            store[i] = {
                "master": f.master.detach().cpu(),
                "exp_avg": f.exp_avg.detach().cpu(),
                "exp_avg_sq": f.exp_avg_sq.detach().cpu(),
                "mode": f.mode,
            }
    payload = {
        "iteration": iteration,
        "blocks": blocks,
        "expert_blocks": expert_blocks,
        "optimizer": {"step_count": opt.step_count},
        "scheduler": sched.state_dict() if sched is not None else None,
        "rng": {"torch": torch.get_rng_state(),
                "cuda": (torch.cuda.get_rng_state()
                         if torch.cuda.is_available() else None)},
        "world_size": world,
        # fault-attribution state travels with the run (reference
        # rerun_state_machine.py:871-902 persists into checkpoints)
        "rerun_state_machine": (rerun_state_machine.state_dict()
                                if rerun_state_machine is not None else None),
    }
    torch.save(payload, os.path.join(it_dir, f"rank_{rank:05d}.pt"))
    if rank == 0:
        with open(os.path.join(it_dir, "hybrid_parallel_config.json"), "w") as f:
            json.dump(model.plan.to_config_dict(), f, indent=2)
        with open(os.path.join(save_dir,
                               "latest_checkpointed_iteration.txt"), "w") as f:
            f.write(str(iteration))
    if dist.is_initialized():
        dist.barrier()
    return it_dir


def latest_iteration(load_dir: str) -> Optional[int]:
    p = os.path.join(load_dir, "latest_checkpointed_iteration.txt")
    if not os.path.exists(p):
        return None
    with open(p) as f:
        return int(f.read().strip())


def load_distributed_checkpoint(model, opt, sched, cfg,
                                load_dir: Optional[str] = None,
                                iteration: Optional[int] = None,
                                rerun_state_machine=None) -> int:
    load_dir = load_dir or cfg.ckpt.load
    if iteration is None:
        iteration = (cfg.ckpt.load_iteration
                     or latest_iteration(load_dir))
        assert iteration, f"no checkpoint in {load_dir}"
    rank, world = _rank_world()
    it_dir = os.path.join(load_dir, f"iter_{iteration:07d}")

    with open(os.path.join(it_dir, "hybrid_parallel_config.json")) as f:
        saved_plan = json.load(f)
    cur_plan = model.plan.to_config_dict()
    for key in ("pp_deg", "pp_division", "tp_sizes_enc", "cp_sizes_enc",
                "dp_types_enc", "use_sp", "checkpoint_flags", "vtp"):
        assert str(saved_plan.get(key)) == str(cur_plan.get(key)), \
            (f"checkpoint plan mismatch on {key}: saved "
             f"{saved_plan.get(key)} vs current {cur_plan.get(key)} — "
             "distributed checkpoints resume on the SAME plan (convert via "
             "canonical/HF format to change plans)")

    payload = torch.load(os.path.join(it_dir, f"rank_{rank:05d}.pt"),
                         map_location="cpu", weights_only=False)
    assert payload["world_size"] == world, \
        f"world size changed: {payload['world_size']} -> {world}"
    with torch.no_grad():
        for i, blk in enumerate(model.stage_model.blocks):
            for f, store in ((blk.flat, payload["blocks"]),
                             (getattr(blk, "flat_expert", None),
                              payload.get("expert_blocks", {}))):
                if f is None or i not in store:
                    continue
                st = store[i]
                f.master.copy_(st["master"].to(f.master.device))
                f.exp_avg.copy_(st["exp_avg"].to(f.exp_avg.device))
                f.exp_avg_sq.copy_(st["exp_avg_sq"].to(f.exp_avg_sq.device))
                f.apply_master_to_params()
    if opt is not None:
        opt.step_count = payload["optimizer"]["step_count"]
    if sched is not None and payload["scheduler"] is not None:
        sched.load_state_dict(payload["scheduler"])
    if rerun_state_machine is not None \
            and payload.get("rerun_state_machine") is not None:
        rerun_state_machine.load_state_dict(payload["rerun_state_machine"])
    if payload["rng"]["torch"] is not None:
        torch.set_rng_state(payload["rng"]["torch"])
    if payload["rng"]["cuda"] is not None and torch.cuda.is_available():
        torch.cuda.set_rng_state(payload["rng"]["cuda"])
    if dist.is_initialized():
        dist.barrier()
    return iteration
