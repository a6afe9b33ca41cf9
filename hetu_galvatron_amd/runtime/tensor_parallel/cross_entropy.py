"""Vocab-parallel cross entropy over a TP-sharded vocabulary.

Reference: galvatron/core/runtime/tensor_parallel/triton_cross_entropy.py
(Triton tiled kernels + host all-reduces) and transformer/fused_kernels.py:258-430
(torch-ops version).  Here: the tiled shard-local passes are CDNA4 HIP
kernels (ops/csrc/cross_entropy.hip) — max / sum-exp+target-pick / in-place
softmax-grad — with three [n]-shaped fp32 all-reduces (max, sumexp, target
logit) over the TP group between them; CPU falls back to the torch
reference.  The backward writes the gradient IN PLACE over the logits
buffer: at Llama-3's 128k vocab the logits tensor dominates the lm-head's
activation memory and MI355X's 288 GB still appreciates not duplicating it.
"""
from __future__ import annotations


import torch
import torch.distributed as dist

from ...ops import reference_ops as ref
from ...ops._ext import get_ext, use_native
from .mappings import group_rank, group_size


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, group):
        """logits: [n, v_local] (bf16/fp32); target: [n] global ids.
        Returns per-token loss [n] fp32."""
        ws = group_size(group)
        rank = group_rank(group)
        v_local = logits.shape[-1]
        vocab_start = rank * v_local
        vocab_end = vocab_start + v_local

        if use_native(logits):
            ext = get_ext(False)
            local_max = ext.ce_max(logits)
        else:
            local_max = logits.float().max(dim=-1).values
        gmax = local_max
        if ws > 1:
            dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)

        if use_native(logits):
            sumexp, tlogit = ext.ce_sum_target(logits, target, gmax, vocab_start)
        else:
            sumexp, tlogit = ref.vocab_ce_fwd_local(logits, target, gmax,
                                                    vocab_start, vocab_end)
        if ws > 1:
            buf = torch.stack([sumexp, tlogit])
            dist.all_reduce(buf, group=group)
            sumexp, tlogit = buf[0], buf[1]
        loss = torch.log(sumexp) - (tlogit - gmax)
        ctx.save_for_backward(logits, target, gmax, sumexp)
        ctx.vocab_start = vocab_start
        ctx.vocab_end = vocab_end
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        logits, target, gmax, sumexp = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        if use_native(logits):
            # in-place: logits buffer becomes dlogits
            dlogits = get_ext(False).ce_bwd(logits, target, gmax, sumexp,
                                            grad_out, ctx.vocab_start)
        else:
            dlogits = ref.vocab_ce_bwd_local(logits, target, gmax, sumexp,
                                             grad_out, ctx.vocab_start,
                                             ctx.vocab_end)
        return dlogits, None, None


def vocab_parallel_cross_entropy(logits: torch.Tensor, target: torch.Tensor,
                                 group=None) -> torch.Tensor:
    """Per-token CE loss over a vocab-sharded logits tensor.

    logits: [..., v_local] flattened to [n, v_local]; target: matching [n].
    """
    shape = target.shape
    n = target.numel()
    out = _VocabParallelCrossEntropy.apply(
        logits.reshape(n, logits.shape[-1]), target.reshape(n), group)
    return out.view(shape)
