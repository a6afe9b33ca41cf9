"""Top-K MoE router.

Reference: galvatron/core/runtime/moe/router.py:22-437 (TopKRouter:
softmax/sigmoid scores, pre/post-softmax topk, aux load-balancing loss,
seq-aux, z-loss, sinkhorn init, aux-loss-free expert-bias updates,
capacity-factor dropping) and moe_utils.py:14-240.
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class MoEAuxLossAutoScaler(torch.autograd.Function):
    """Pass the activation through; inject the aux-loss gradient on backward
    (reference moe_utils.py:166 MoEAuxLossAutoScaler).

    main_loss_backward_scale mirrors the main loss's backward scaling —
    the engine sets it to loss_scale/num_microbatches each step so the
    accumulated aux gradient represents coeff * mean-over-microbatches
    (and survives fp16 scale-unscale)."""

    main_loss_backward_scale: float = 1.0

    @staticmethod
    def forward(ctx, output: torch.Tensor, aux_loss: torch.Tensor):
        ctx.save_for_backward(aux_loss)
        return output

    @staticmethod
    def backward(ctx, grad_output: torch.Tensor):
        (aux_loss,) = ctx.saved_tensors
        scale = MoEAuxLossAutoScaler.main_loss_backward_scale
        return grad_output, torch.full_like(aux_loss, scale)


def attach_aux_loss(x: torch.Tensor, aux_loss: torch.Tensor) -> torch.Tensor:
    return MoEAuxLossAutoScaler.apply(x, aux_loss)


def sinkhorn(cost: torch.Tensor, tol: float = 1e-4,
             iters: int = 8) -> torch.Tensor:
    """Sinkhorn normalization over [n, E] (reference moe_utils.py:130)."""
    cost = torch.exp(cost)
    d0 = torch.ones(cost.size(0), device=cost.device)
    d1 = torch.ones(cost.size(1), device=cost.device)
    eps = 1e-8
    for _ in range(iters):
        d0 = (1.0 / cost.size(0)) / ((cost * d1.unsqueeze(0)).sum(1) + eps)
        d1 = (1.0 / cost.size(1)) / ((cost * d0.unsqueeze(1)).sum(0) + eps)
    return cost * d1.unsqueeze(0) * d0.unsqueeze(1)


class TopKRouter(nn.Module):
    def __init__(self, margs, dtype=None):
        super().__init__()
        self.num_experts = margs.num_experts
        self.topk = margs.moe_router_topk
        self.score_function = margs.moe_router_score_function
        self.pre_softmax = margs.moe_router_pre_softmax
        self.aux_loss_coeff = margs.moe_aux_loss_coeff
        self.z_loss_coeff = margs.moe_z_loss_coeff
        self.aux_loss_free = margs.moe_aux_loss_free
        self.bias_update_rate = margs.moe_router_bias_update_rate
        self.capacity_factor = margs.moe_expert_capacity_factor
        self.aux_loss_type = getattr(margs, "moe_aux_loss_type", "aux_loss")
        self.load_balancing_type = getattr(
            margs, "moe_router_load_balancing_type", "none")
        # train.deterministic_mode (reference: deterministic router topk):
        # stable argsort instead of torch.topk so expert ties break by
        # index on every backend/run; set by the builder
        self.deterministic = False
        self.num_groups = getattr(margs, "moe_router_num_groups", None)
        self.group_topk = getattr(margs, "moe_router_group_topk", None)
        if self.num_groups:
            assert self.num_experts % self.num_groups == 0
            assert self.group_topk and self.group_topk <= self.num_groups
        # fp32 router weight (routing numerics, reference router.py:70)
        self.weight = nn.Parameter(
            torch.empty(self.num_experts, margs.hidden_size,
                        dtype=torch.float32))
        nn.init.normal_(self.weight, 0.0,
                        getattr(margs, "init_method_std", 0.02))
        if self.aux_loss_free:
            self.register_buffer(
                "expert_bias", torch.zeros(self.num_experts,
                                           dtype=torch.float32))
        else:
            self.expert_bias = None

    def forward(self, x: torch.Tensor, seq_len: int = None
                ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """x: [n, h] -> (probs [n,k], indices [n,k] long, logits [n,E]).
        seq_len: rows per sequence in x (n = seq_len*b, seq-major), needed
        only for moe_aux_loss_type=seq_aux_loss."""
        # routing math in fp32 (the flat-param engine may hold the
        # weight in bf16; cast per-call, grads flow through the cast)
        logits = F.linear(x.float(), self.weight.float())  # [n, E]
        n, E = logits.shape

        if self.score_function == "sigmoid":
            scores = torch.sigmoid(logits)
        elif self.pre_softmax:
            scores = torch.softmax(logits, dim=-1)
        else:
            scores = logits

        sel = scores
        if self.load_balancing_type == "sinkhorn" and self.training:
            # sinkhorn routing (reference router.py:140): expert choice
            # comes from the balanced sinkhorn matrix, probs from the
            # raw scores at those experts (gradient flows through scores)
            with torch.no_grad():
                sel = sinkhorn(logits.detach().float())
        if self.expert_bias is not None:
            sel = scores + self.expert_bias.unsqueeze(0)
        if self.num_groups:
            # group-limited (node-limited) routing, DeepSeek-style
            # (reference moe_utils.py:372 group_limited_topk): score each
            # expert group by its top-2 sum, keep the best group_topk
            # groups, mask the rest before the expert topk
            G = self.num_groups
            gs = sel.view(n, G, E // G)
            group_score = gs.topk(min(2, E // G), dim=-1).values.sum(-1)
            keep = group_score.topk(self.group_topk, dim=-1).indices
            gmask = torch.zeros(n, G, device=sel.device,
                                dtype=torch.bool).scatter_(1, keep, True)
            sel = gs.masked_fill(~gmask.unsqueeze(-1),
                                 float("-inf")).view(n, E)
        if self.deterministic:
            order = torch.argsort(sel, dim=-1, descending=True,
                                  stable=True)
            idx = order[:, : self.topk]
            top_vals = torch.gather(sel, 1, idx)
        else:
            top_vals, idx = torch.topk(sel, self.topk, dim=-1)
        gathered = torch.gather(scores, 1, idx)

        if self.score_function == "sigmoid":
            probs = gathered / (gathered.sum(-1, keepdim=True) + 1e-20)
        elif self.pre_softmax:
            probs = gathered / (gathered.sum(-1, keepdim=True) + 1e-20) \
                if self.topk > 1 else gathered
        else:
            probs = torch.softmax(gathered, dim=-1)

        if self.capacity_factor:
            # capacity dropping (reference moe_utils.py:147
            # topk_softmax_with_capacity): tokens beyond
            # ceil(n*topk/E * cf) per expert, in token order, get prob 0
            # (they still travel through the dispatcher — pad-to-capacity
            # buffers are a v2 item — but contribute nothing)
            import math
            cap = int(math.ceil(n * self.topk / E * self.capacity_factor))
            with torch.no_grad():
                flat_idx = idx.flatten()
                onehot = F.one_hot(flat_idx, E)
                rank_in_expert = onehot.cumsum(0).gather(
                    1, flat_idx.unsqueeze(1)).squeeze(1)  # 1-based
                keep_tok = (rank_in_expert <= cap).view(n, self.topk)
            probs = probs * keep_tok.to(probs.dtype)

        aux = logits.new_zeros(())
        if self.aux_loss_coeff > 0 and not self.aux_loss_free \
                and self.training:
            with torch.no_grad():
                mask = torch.zeros_like(logits).scatter_(
                    1, idx, 1.0)
            P_tok = torch.softmax(logits, dim=-1)
            if self.aux_loss_type == "seq_aux_loss" and seq_len is not None \
                    and n % seq_len == 0:
                # per-sequence balance, averaged over the batch
                # (reference moe_utils.py:62 sequence_load_balancing)
                b = n // seq_len
                f = mask.view(seq_len, b, E).mean(0) * E / self.topk
                P = P_tok.view(seq_len, b, E).mean(0)
                aux = aux + self.aux_loss_coeff * (f * P).sum(-1).mean()
            else:
                # load-balancing loss: E * sum_e f_e * P_e
                # (reference moe_utils.py:14)
                f = mask.mean(0) * E / self.topk
                P = P_tok.mean(0)
                aux = aux + self.aux_loss_coeff * (f * P).sum()
        if self.z_loss_coeff > 0 and self.training:
            aux = aux + self.z_loss_coeff * \
                torch.logsumexp(logits, dim=-1).square().mean()
        if self.aux_loss_free and self.training:
            with torch.no_grad():
                load = torch.zeros(E, device=logits.device)
                load.scatter_add_(0, idx.flatten(),
                                  torch.ones_like(idx.flatten(),
                                                  dtype=load.dtype))
                err = load.mean() - load  # positive = underloaded
                self.expert_bias += self.bias_update_rate * torch.sign(err)
        return probs.to(x.dtype), idx, aux
