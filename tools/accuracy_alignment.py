#!/usr/bin/env python3
"""Accuracy-alignment harness: loss-curve comparison between two configs.

Reference: galvatron/scripts/accuracy_alignment/ (baseline-vs-test loss
curves uploaded to wandb).  Runs the SAME model from the SAME initial
weights on the SAME batches under two parallel configurations and reports
per-step loss deltas; any hybrid-parallel plan must reproduce the
1-process loss curve within bf16 tolerance.

  python tools/accuracy_alignment.py --model tiny-llama --iters 10 \\
      --test-overrides parallel.global_tp_deg=2 --world 2 --out align.json
(world > 1 spawns gloo/CPU or RCCL/GPU subprocesses like the test harness.)
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run_config(model_name, iters, bsz, overrides, state_path=None,
               dump_state=None):
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)
    from hetu_galvatron_amd.runtime.checkpoint import (
        canonical_state_from_stage, load_full_state)

    cfg = load_config(
        base={"model": {"model_name": model_name},
              "train": {"global_train_batch_size": bsz, "train_iters": iters,
                        "lr": 1e-3, "lr_decay_style": "constant"}},
        overrides=overrides)
    torch.manual_seed(cfg.train.seed)
    model = GalvatronModel(cfg)
    if state_path:
        state = torch.load(state_path, weights_only=True)
        load_full_state(model.stage_model, state, cfg.model)
    if dump_state:
        torch.save(canonical_state_from_stage(model.stage_model), dump_state)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    it = get_train_iterator(cfg, device)
    losses = []
    for _ in range(iters):
        opt.zero_grad()
        stats = model.forward_backward(next(it))
        opt.step()
        sched.step()
        losses.append(model.global_loss(stats))
    return losses


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny-llama")
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--bsz", type=int, default=4)
    ap.add_argument("--baseline-overrides", nargs="*", default=[])
    ap.add_argument("--test-overrides", nargs="*", default=[])
    ap.add_argument("--tol", type=float, default=0.03)
    ap.add_argument("--out", default="accuracy_alignment.json")
    args = ap.parse_args()

    state = "/tmp/align_state.pt"
    base = run_config(args.model, args.iters, args.bsz,
                      args.baseline_overrides, dump_state=state)
    test = run_config(args.model, args.iters, args.bsz,
                      args.test_overrides, state_path=state)
    deltas = [abs(a - b) for a, b in zip(base, test)]
    report = {"model": args.model, "iters": args.iters,
              "baseline_losses": base, "test_losses": test,
              "max_abs_delta": max(deltas), "tol": args.tol,
              "aligned": max(deltas) <= args.tol}
    with open(args.out, "w") as f:
        json.dump(report, f, indent=2)
    print(json.dumps({k: report[k] for k in
                      ("max_abs_delta", "aligned")}, indent=2))
    sys.exit(0 if report["aligned"] else 1)


if __name__ == "__main__":
    main()
