"""Replicated-parameter consistency checks (determinism guard).

Reference role: the reference's `run_realtime_tests` "test_mode"
(attention.py:805-875) compares TP-replicated params across ranks at
runtime to catch desynchronization (missed grad sync, nondeterministic
kernels).  Here a framework-wide sweep: every parameter that is
REPLICATED across a group must be bit-identical on all its ranks —
  * tp_replicated-tagged params across the layer's tp group,
  * every param across its ZeRO sdp group replicas (ddp and zero2 —
    zero2's post-step allgather re-replicates the bf16 params),
  * expert params across the edp group.
Enable per-interval via train.check_weight_consistency_interval.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist


def _group_mismatch(t: torch.Tensor, group) -> bool:
    """True if `t` differs across `group` (fingerprint all-gather)."""
    fp = torch.stack([t.detach().float().sum(),
                      t.detach().float().norm(),
                      t.detach().flatten()[:1].float().squeeze()
                      if t.numel() else t.new_zeros(())])
    ws = dist.get_world_size(group)
    outs = [torch.empty_like(fp) for _ in range(ws)]
    dist.all_gather(outs, fp, group=group)
    return any(not torch.equal(outs[0], o) for o in outs[1:])


def check_param_consistency(stage_model) -> List[str]:
    """Returns names of parameters whose replicas diverged."""
    bad: List[str] = []
    if not dist.is_initialized():
        return bad
    for bi, blk in enumerate(stage_model.blocks):
        g = blk.groups
        tp = g.tp_group if not g.strategy.use_ulysses else None
        sdp = g.sdp_group
        edp = getattr(g, "edp_group", None)
        for name, p in blk.inner.named_parameters():
            expert = getattr(p, "expert_parallel", False)
            # replication domain for this param
            if expert:
                grp = edp.group if (edp is not None and edp.size > 1) \
                    else None
            elif blk.flat is not None \
                    and blk.flat.mode in ("ddp", "zero2") \
                    and sdp is not None and sdp.size > 1:
                # ddp: fully replicated; zero2: bf16 params re-replicated
                # by the post-step allgather — either way replicas must
                # be identical between steps
                grp = sdp.group
            else:
                grp = None
            if grp is not None and _group_mismatch(p, grp):
                bad.append(f"block{bi}.{name}[dp]")
            if not expert and tp is not None and tp.size > 1 \
                    and getattr(p, "tp_replicated", False) \
                    and _group_mismatch(p, tp.group):
                bad.append(f"block{bi}.{name}[tp]")
    return bad
