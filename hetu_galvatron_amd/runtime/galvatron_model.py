"""Top-level hybrid-parallel model: StageModel + PipelineEngine + step API.

Reference: galvatron/core/runtime/hybrid_parallel_model.py:50-107
(GalvatronModel.forward_backward dispatching the schedules).
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..config import GalvatronConfig, HybridParallelPlan
from .models.builder import build_hybrid_parallel_model
from .pipeline.engine import PipelineEngine, StepStats


def resolve_plan(cfg: GalvatronConfig, world_size: int) -> HybridParallelPlan:
    """Searched-plan JSON if given, else GLOBAL-mode uniform plan
    (reference: hybrid_parallel_config.py:18 get_hybrid_parallel_configs_api)."""
    p = cfg.parallel
    n_layers = cfg.model.num_hidden_layers
    if cfg.model.model_type == "t5":
        n_layers += cfg.model.num_decoder_layers or cfg.model.num_hidden_layers
    if p.galvatron_config_path:
        plan = HybridParallelPlan.load(p.galvatron_config_path)
    else:
        dp_type = "zero3" if p.sdp else p.default_dp_type
        chunks = p.chunks
        if chunks is None or chunks < 1:
            # auto heuristic (reference: hybrid_parallel_config.py:359
            # get_chunks): enough microbatches to fill the pipeline, at
            # least 2 per-stage for grad-accum overlap, bounded by the
            # per-dp-rank batch
            dp = world_size // max(p.pp_deg * p.global_tp_deg
                                   * p.global_cp_deg, 1)
            per_dp = max(cfg.train.global_train_batch_size // max(dp, 1), 1)
            chunks = min(max(2 * p.pp_deg, 2), per_dp) if p.pp_deg > 1 \
                else min(2, per_dp)
        plan = HybridParallelPlan.uniform(
            num_layers=n_layers, world_size=world_size,
            pp=p.pp_deg, tp=p.global_tp_deg, cp=p.global_cp_deg,
            use_sp=p.use_ulysses, dp_type=dp_type,
            checkpoint=bool(p.global_checkpoint),
            chunks=chunks, global_bsz=cfg.train.global_train_batch_size,
            pipeline_type=p.pipeline_type, vtp=p.vocab_tp,
            vsp=bool(p.vocab_sp), ep=p.global_ep_deg)
    if plan.global_bsz != cfg.train.global_train_batch_size:
        plan.global_bsz = cfg.train.global_train_batch_size
    return plan


class GalvatronModel(nn.Module):
    """Hybrid-parallel causal LM ready for forward_backward steps."""

    def __init__(self, cfg: GalvatronConfig,
                 plan: Optional[HybridParallelPlan] = None,
                 device: Optional[torch.device] = None):
        super().__init__()
        world = dist.get_world_size() if dist.is_initialized() else 1
        self.cfg = cfg
        self.plan = plan or resolve_plan(cfg, world)
        self.stage_model = build_hybrid_parallel_model(cfg, self.plan, device)
        act_dtype = {"bf16": torch.bfloat16,
                     "fp16": torch.float16}.get(
            cfg.parallel.mixed_precision, torch.float32)
        self.engine = PipelineEngine(
            self.stage_model, cfg.model.hidden_size,
            pipeline_type=self.plan.pipeline_type,
            overlap_grad_reduce=cfg.parallel.overlap_grad_reduce,
            act_dtype=act_dtype)
        self.chunks = max(self.plan.chunks, 1)

    @property
    def blocks(self):
        return self.stage_model.blocks

    def forward_backward(self, ctx: Dict, chunks: Optional[int] = None) -> StepStats:
        scaler = getattr(self.stage_model, "loss_scaler", None)
        if scaler is not None:
            self.engine.loss_scale = scaler.scale
        return self.engine.forward_backward(ctx, chunks or self.chunks)

    def evaluate(self, ctx):
        """Forward-only validation pass -> StepStats."""
        return self.engine.evaluate(ctx, self.plan.chunks)

    def global_loss(self, stats: StepStats, device=None) -> float:
        """Token-weighted mean loss across all ranks (handles vtp row
        duplication by weighting with per-rank token counts)."""
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return stats.loss
        dev = device or self.engine.device
        t = torch.tensor([stats.loss_sum, stats.token_count],
                         dtype=torch.float64,
                         device=dev if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t)
        return float(t[0] / max(t[1].item(), 1.0))
