"""Distributed optimizer over the ZeRO blocks' master partitions.

Reference: galvatron/core/runtime/optimizer/utils.py:8-70 (apex FusedAdam ->
here the HIP multi-tensor AdamW kernel ops/csrc/adamw.hip) and
clip_grads.py:11-194 (grad-norm + clip; the MoE expert-grad rescale hooks in
here once EP lands).

Each FlatParamBlock owns a master fp32 partition + adam moments; the fused
kernel updates master/m/v AND writes the bf16 model shard in one pass
(saving one full HBM sweep per step on MI355X's 8 TB/s HBM3E).  Grad norm:
every rank sums squares over elements it OWNS (each model element counted
once world-wide), one scalar all-reduce, then a multi-tensor scale.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from ...ops import reference_ops as ref
from ...ops._ext import get_ext, native_available
from ..zero import FlatParamBlock


class LossScaler:
    """Megatron-style dynamic loss scaling for fp16 (reference:
    optimizer mixed-precision wrapper).  bf16 needs none (same exponent
    range as fp32), which is why bf16 is this framework's default."""

    def __init__(self, init_scale: float = 65536.0, window: int = 1000,
                 min_scale: float = 1.0, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5):
        self.scale = float(init_scale)
        self.window = window
        self.min_scale = min_scale
        self.growth = growth_factor
        self.backoff = backoff_factor
        self._good_steps = 0

    def update(self, found_inf: bool) -> None:
        if found_inf:
            self.scale = max(self.scale * self.backoff, self.min_scale)
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.window:
                self.scale *= self.growth
                self._good_steps = 0

    def state_dict(self) -> dict:
        return {"scale": self.scale, "good_steps": self._good_steps}

    def load_state_dict(self, sd: dict) -> None:
        self.scale = sd["scale"]
        self._good_steps = sd["good_steps"]


class GalvatronOptimizer:
    def __init__(self, blocks: List[FlatParamBlock], lr: float = 1e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.01, clip_grad: float = 1.0,
                 use_fused: bool = True,
                 loss_scaler: Optional[LossScaler] = None):
        self.blocks = [b for b in blocks if b is not None and b.total > 0]
        self.loss_scaler = loss_scaler
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.clip_grad = clip_grad
        self.use_fused = use_fused
        self.step_count = 0
        self.last_grad_norm: Optional[float] = None

    # -- grad norm + clip ---------------------------------------------------

    def _global_grad_norm(self) -> float:
        dev = self.blocks[0].device if self.blocks else torch.device("cpu")
        if (native_available() and dev.type == "cuda" and self.blocks
                and all(b.grad_shard is not None for b in self.blocks)):
            # one fused multi-tensor Σg² launch over every owned shard
            # (the eager per-block pow+reduce chain was ~2 launches per
            # block per step), plus tiny eager corrections for
            # tp-replicated segments
            views = [b.owned_grad_view() for b in self.blocks]
            total = get_ext(False).multi_sumsq(views)[0]
            for b in self.blocks:
                c = b.sumsq_tp_correction()
                if c is not None:
                    total = total + c
        else:
            total = torch.zeros((), dtype=torch.float32, device=dev)
            for b in self.blocks:
                total = total + b.grad_sumsq_owned()
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(total)
        return float(total.sqrt().item())

    def clip_gradients(self) -> float:
        norm = self._global_grad_norm()
        self.last_grad_norm = norm
        if self.clip_grad and self.clip_grad > 0 and norm > self.clip_grad:
            scale = self.clip_grad / (norm + 1e-6)
            for b in self.blocks:
                b.scale_grads(scale)
        return norm

    # -- step ---------------------------------------------------------------

    def _block_tensors(self, b: FlatParamBlock):
        master, grad = b.owned_master_and_grad()
        if b.mode == "ddp":
            bf16_out = b.flat_param
        elif b.mode == "zero2":
            bf16_out = b.flat_param[b._shard_slice()]
        else:
            bf16_out = b.param_shard
        return master, grad, b.exp_avg, b.exp_avg_sq, bf16_out

    def step(self) -> float:
        use_native = (self.use_fused and native_available()
                      and self.blocks and self.blocks[0].device.type == "cuda")
        inv_ls = 1.0 / self.loss_scaler.scale if self.loss_scaler else 1.0
        norm = self._global_grad_norm() * inv_ls  # grads carry the loss scale
        self.last_grad_norm = norm
        if self.loss_scaler is not None:
            import math as _math
            found_inf = not _math.isfinite(norm)
            self.loss_scaler.update(found_inf)
            if found_inf:
                # global decision (the norm was all-reduced): skip the step,
                # back off the scale (reference fp16 semantics)
                self.last_grad_norm = float("nan")
                return float("nan")
        scale = inv_ls
        if self.clip_grad and self.clip_grad > 0 and norm > self.clip_grad:
            scale = inv_ls * self.clip_grad / (norm + 1e-6)
        if scale != 1.0 and not use_native:
            # fallback path applies unscale+clip as a separate pass; the
            # native path folds it into the adam kernel (gscale)
            for b in self.blocks:
                b.scale_grads(scale)
        self.step_count += 1
        masters, grads, ms, vs, outs = [], [], [], [], []
        for b in self.blocks:
            t = self._block_tensors(b)
            masters.append(t[0]); grads.append(t[1]); ms.append(t[2])
            vs.append(t[3]); outs.append(t[4])
        if use_native:
            get_ext(False).fused_adamw(
                masters, grads, ms, vs, outs, self.step_count, self.lr,
                self.betas[0], self.betas[1], self.eps, self.weight_decay,
                scale)  # scale = 1/loss_scale * clip factor
        else:
            ref.adamw_step(outs, grads, ms, vs, masters, self.step_count,
                           self.lr, self.betas[0], self.betas[1], self.eps,
                           self.weight_decay)
        for b in self.blocks:
            if b.mode == "zero2":
                b.apply_master_to_params_post_step()
        return norm

    def zero_grad(self) -> None:
        for b in self.blocks:
            b.zero_grad()

    # -- state --------------------------------------------------------------

    def state_dict(self) -> dict:
        return {
            "step_count": self.step_count,
            "blocks": [
                {"master": b.master, "exp_avg": b.exp_avg,
                 "exp_avg_sq": b.exp_avg_sq}
                for b in self.blocks
            ],
        }

    def load_state_dict(self, sd: dict) -> None:
        self.step_count = sd["step_count"]
        for b, bs in zip(self.blocks, sd["blocks"]):
            b.master.copy_(bs["master"].to(b.master.device))
            b.exp_avg.copy_(bs["exp_avg"].to(b.master.device))
            b.exp_avg_sq.copy_(bs["exp_avg_sq"].to(b.master.device))
            b.apply_master_to_params()


def get_optimizer_and_param_scheduler(stage_model, cfg):
    """reference: optimizer/utils.py:44."""
    from .scheduler import OptimizerParamScheduler

    blocks = []
    for blk in stage_model.blocks:
        if getattr(blk, "flat_expert", None) is not None:
            blocks.append(blk.flat_expert)
        if blk.flat is not None:
            blocks.append(blk.flat)
    t = cfg.train
    scaler = None
    if cfg.parallel.mixed_precision == "fp16":
        scaler = LossScaler(cfg.parallel.loss_scale_init,
                            cfg.parallel.loss_scale_window,
                            cfg.parallel.min_loss_scale)
        stage_model.loss_scaler = scaler
    opt = GalvatronOptimizer(
        blocks, lr=t.lr, betas=(t.adam_beta1, t.adam_beta2), eps=t.adam_eps,
        weight_decay=t.adam_weight_decay, clip_grad=t.clip_grad,
        use_fused=t.use_fused_adam, loss_scaler=scaler)
    sched = OptimizerParamScheduler(
        opt, max_lr=t.lr, min_lr=t.min_lr, warmup_steps=t.lr_warmup_iters,
        decay_steps=t.lr_decay_iters or t.train_iters,
        decay_style=t.lr_decay_style, wsd_decay_steps=t.lr_wsd_decay_iters,
        warmup_init_lr=t.lr_warmup_init)
    return opt, sched
