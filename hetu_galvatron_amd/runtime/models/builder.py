"""Declarative model assembly: arch list -> per-stage LayerBlocks.

Reference: galvatron/core/runtime/models/builder.py:42-207 (arch list,
MODULE_REGISTRY), hybrid_parallel_model.py:107 (6-step assembly pipeline).
Steps here: plan -> strategies -> comm groups (collective) -> stage modules
-> LayerBlock wrap (relocation + ZeRO + activation ckpt).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.utils.checkpoint as torch_ckpt

from ...config import GalvatronConfig, HybridParallelPlan, even_pp_division
from ...core.comm_groups import (
    CommGroup,
    CommGroupCache,
    LayerCommGroups,
    gen_embedding_group,
    gen_layer_comm_groups,
    pp_stage_of_rank,
)
from ..redistribute import redistribute
from ..zero import FlatParamBlock, _PostBackwardHook, _PreBackwardGather
from .modules import (
    GalvatronCausalLMHead, GalvatronDecoderLayer, GalvatronEmbedding,
    GalvatronFinalNorm,
)
from .encdec_modules import (GalvatronDecoderLayerX, GalvatronEncDecBridge,
                             GalvatronEncoderLayer)
from .moe_modules import GalvatronMoEDecoderLayer


def build_causal_lm_arch(num_layers: int) -> List[str]:
    """reference: builder.py:111 build_causal_lm_arch."""
    return ["embedding"] + ["decoder"] * num_layers + ["final_norm", "lm_head"]


def build_enc_dec_arch(num_enc: int, num_dec: int) -> List[str]:
    """t5-style: plan layers map 1:1 to encoder then decoder layers."""
    return (["embedding"] + ["encoder"] * num_enc + ["encdec_bridge"] +
            ["decoder_x"] * num_dec + ["final_norm", "lm_head"])


class LayerBlock(nn.Module):
    """One pipeline cell: input relocation + ZeRO-managed inner module +
    optional activation checkpointing.
    Reference: parallel.py:272 Module_with_relocation + wrap_modules_*."""

    def __init__(self, inner: nn.Module, groups: LayerCommGroups, kind: str,
                 checkpoint: bool = False,
                 prev_groups: Optional[LayerCommGroups] = None):
        super().__init__()
        self.inner = inner
        self.groups = groups
        self.kind = kind  # embedding | decoder | final_norm | lm_head
        self.checkpoint = checkpoint
        self.prev_groups = prev_groups
        self.flat: Optional[FlatParamBlock] = None
        self.flat_expert: Optional[FlatParamBlock] = None
        # neighbor links for zero3 prefetch (set by build_hybrid_parallel_model)
        self.next_blk: Optional["LayerBlock"] = None
        self.prev_blk: Optional["LayerBlock"] = None

    def setup_zero(self, mode: str, param_dtype: torch.dtype, device,
                   reduce_in_fp32: bool = False,
                   reduce_each_microbatch: bool = False) -> None:
        tp_group = None
        if not self.groups.strategy.use_ulysses:
            tp_group = self.groups.tp_group
        # MoE layers: expert params reduce over EDP (dp-of-experts), the rest
        # over the sdp group (reference parallel.py MoE double-wrap)
        if self.groups.ep_group is not None and self.groups.ep_group.size > 1:
            self.flat_expert = FlatParamBlock(
                self.inner, mode, self.groups.edp_group, tp_group=None,
                param_dtype=param_dtype, device=device,
                param_filter=lambda p: getattr(p, "expert_parallel", False),
                reduce_in_fp32=reduce_in_fp32,
                reduce_each_microbatch=reduce_each_microbatch)
        self.flat = FlatParamBlock(self.inner, mode, self.groups.sdp_group,
                                   tp_group=tp_group, param_dtype=param_dtype,
                                   device=device,
                                   reduce_in_fp32=reduce_in_fp32,
                                   reduce_each_microbatch=reduce_each_microbatch)

    def _inner_forward(self, x, ctx):
        if self.kind == "embedding":
            return self.inner(ctx)
        return self.inner(x, ctx)

    def forward(self, x: Optional[torch.Tensor], ctx: Dict) -> torch.Tensor:
        if x is not None and self.prev_groups is not None:
            x = redistribute(x, self.prev_groups, self.groups, ctx["batch_size"])
        if self.flat is not None:
            self.flat.gather_params()
        if self.flat_expert is not None:
            self.flat_expert.gather_params()
        # zero3 prefetch: kick the NEXT block's param all-gather so it
        # overlaps this block's forward compute
        if self.next_blk is not None:
            if self.next_blk.flat is not None:
                self.next_blk.flat.prefetch_gather()
            if self.next_blk.flat_expert is not None:
                self.next_blk.flat_expert.prefetch_gather()
        if x is not None and self.flat is not None and x.requires_grad:
            x = _PostBackwardHook.apply(self.flat, x)
        if x is not None and self.flat_expert is not None and x.requires_grad:
            x = _PostBackwardHook.apply(self.flat_expert, x)
        if self.checkpoint and self.kind in ("decoder", "encoder") \
                and torch.is_grad_enabled():
            out = torch_ckpt.checkpoint(
                lambda t: self._inner_forward(t, ctx), x, use_reentrant=False)
        else:
            out = self._inner_forward(x, ctx)
        prev_flat = self.prev_blk.flat if self.prev_blk is not None else None
        prev_exp = self.prev_blk.flat_expert if self.prev_blk is not None \
            else None
        if self.flat_expert is not None:
            if out.requires_grad:
                out = _PreBackwardGather.apply(self.flat_expert, out, prev_exp)
            self.flat_expert.reshard_params()
        if self.flat is not None:
            if out.requires_grad:
                # while this block's backward runs, prefetch the previous
                # block (the next one needed in backward order)
                out = _PreBackwardGather.apply(self.flat, out, prev_flat)
            self.flat.reshard_params()
        return out

    def finalize_backward(self) -> None:
        """Idempotent cleanup after a full backward (embedding has no input
        sentinel; zero3 reshard + auto-sync kick happen here)."""
        if self.flat is not None:
            self.flat.post_backward()
        if self.flat_expert is not None:
            self.flat_expert.post_backward()


@dataclass
class StageModel:
    """Everything one rank needs to run its pipeline stage."""

    blocks: List[LayerBlock]
    stage: int
    pp_deg: int
    world_size: int
    rank: int
    plan: HybridParallelPlan
    embed_comm_group: Optional[CommGroup] = None
    tied_embedding_pair: Optional[Tuple[LayerBlock, LayerBlock]] = None
    cache: Optional[CommGroupCache] = None
    # layouts at the stage boundaries (for p2p shapes + boundary relocation)
    recv_layout: Optional[LayerCommGroups] = None   # prev stage's last layer
    send_layout: Optional[LayerCommGroups] = None   # my last layer
    # t5 pipeline boundaries (see build_hybrid_parallel_model)
    recv_is_encoder: bool = False
    recv_carries_memory: bool = False
    send_carries_memory: bool = False

    @property
    def is_first(self) -> bool:
        return self.stage == 0

    @property
    def is_last(self) -> bool:
        return self.stage == self.pp_deg - 1

    def parameters(self):
        for b in self.blocks:
            yield from b.parameters()


def layers_of_stage(division: List[int], stage: int) -> Tuple[int, int]:
    start = sum(division[:stage])
    return start, start + division[stage]


def build_hybrid_parallel_model(cfg: GalvatronConfig, plan: HybridParallelPlan,
                                device: Optional[torch.device] = None
                                ) -> StageModel:
    """Construct this rank's stage of the hybrid-parallel model.

    Collective: every rank must call with the identical plan
    (reference: construct_hybrid_parallel_model_api hybrid_parallel_model.py:107).
    """
    world = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    if device is None:
        device = torch.device("cuda", torch.cuda.current_device()) \
            if torch.cuda.is_available() else torch.device("cpu")
    plan.validate(world)
    margs = cfg.model
    n_layers = plan.num_layers
    pp = plan.pp_deg
    division = plan.pp_division or even_pp_division(n_layers, pp)
    my_stage = pp_stage_of_rank(rank, world, pp)
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16}.get(
        cfg.parallel.mixed_precision, torch.float32)

    strategies = [plan.layer(i, world) for i in range(n_layers)]
    vocab_strat = plan.vocab_strategy(world)
    all_strats = [vocab_strat] + strategies
    groups_list, cache = gen_layer_comm_groups(all_strats, world, rank)
    vg, layer_groups = groups_list[0], groups_list[1:]
    embed_group = gen_embedding_group(world, pp, cache, rank)

    lo, hi = layers_of_stage(division, my_stage)
    blocks: List[LayerBlock] = []
    prev: Optional[LayerCommGroups] = None
    emb_block: Optional[LayerBlock] = None
    head_block: Optional[LayerBlock] = None

    torch.manual_seed(cfg.train.seed)  # deterministic per-rank module init
    if my_stage == 0:
        emb = GalvatronEmbedding(
            margs, vg, dtype=dtype,
            ids_key="enc_input_ids" if margs.model_type == "t5"
            else "input_ids")
        emb_block = LayerBlock(emb, vg, "embedding")
        blocks.append(emb_block)
        prev = vg
    else:
        # incoming activation carries the PREVIOUS stage's last-layer layout
        prev_strategy = strategies[lo - 1]
        prev = gen_layer_comm_groups([prev_strategy], world, rank, cache)[0][0]

    is_moe = margs.model_type.startswith("moe") and margs.num_experts > 0
    is_encdec = margs.model_type == "t5"
    n_enc = margs.num_hidden_layers if is_encdec else 0
    if is_encdec:
        n_dec = margs.num_decoder_layers or margs.num_hidden_layers
        assert n_layers == n_enc + n_dec, \
            f"t5 plan must cover enc+dec layers ({n_enc}+{n_dec}), got {n_layers}" 
    if is_encdec:
        dps = {plan.layer(i, world).dp for i in range(n_layers)}
        dps.add(vocab_strat.dp)
        assert len(dps) == 1, "t5: uniform dp degree required (memory " \
            "replication domain); per-layer tp/zero/ckpt may still vary"
    for i in range(lo, hi):
        lg = layer_groups[i]
        if is_encdec:
            if i < n_enc:
                dec = GalvatronEncoderLayer(margs, lg, layer_idx=i,
                                            dtype=dtype)
                kind = "encoder"
            else:
                dec = GalvatronDecoderLayerX(margs, lg, layer_idx=i,
                                             dtype=dtype)
                kind = "decoder"
            if i == n_enc:  # first decoder layer: bridge precedes it
                bridge = GalvatronEncDecBridge(margs, vg, dtype=dtype)
                bblk = LayerBlock(bridge, vg, "encdec_bridge",
                                  prev_groups=prev)
                blocks.append(bblk)
                prev = vg
        elif is_moe:
            dec = GalvatronMoEDecoderLayer(margs, lg, layer_idx=i, dtype=dtype)
            kind = "decoder"
        else:
            dec = GalvatronDecoderLayer(margs, lg, layer_idx=i, dtype=dtype)
            kind = "decoder"
        blk = LayerBlock(dec, lg, kind,
                         checkpoint=bool(plan.checkpoint_flags[i]),
                         prev_groups=prev)
        blocks.append(blk)
        prev = lg

    if my_stage == pp - 1:
        fn = GalvatronFinalNorm(margs, vg, dtype=dtype)
        blocks.append(LayerBlock(fn, vg, "final_norm", prev_groups=prev))
        head = GalvatronCausalLMHead(margs, vg, dtype=dtype)
        head_block = LayerBlock(head, vg, "lm_head", prev_groups=None)
        if margs.tie_word_embeddings and not margs.untie_embeddings_and_output_weights:
            if pp == 1 and emb_block is not None:
                head.tie_to(emb_block.inner)
        blocks.append(head_block)

    for b in blocks:
        b.inner.to(device)
        mode = b.groups.strategy.dp_type
        b.setup_zero(mode, dtype, device,
                     reduce_in_fp32=cfg.parallel.gradient_reduce_in_fp32,
                     reduce_each_microbatch=
                     cfg.parallel.reduce_grads_each_microbatch)
    if emb_block is not None and head_block is not None:
        # pp-tied embedding/lm-head: the tied grads all-reduce over the
        # embedding group BEFORE the sdp reduction — incompatible with
        # per-microbatch shard-domain accumulation, so those two blocks
        # keep the full accumulator
        for tb in (emb_block, head_block):
            if tb.flat is not None:
                tb.flat.reduce_each_microbatch = False
    for i, b in enumerate(blocks):  # zero3 prefetch neighbor links
        # plain-dict assignment: LayerBlock is an nn.Module and a normal
        # setattr would register the neighbor as a submodule (a cycle)
        b.__dict__["prev_blk"] = blocks[i - 1] if i > 0 else None
        b.__dict__["next_blk"] = blocks[i + 1] if i + 1 < len(blocks) else None

    recv_layout = None
    if my_stage > 0:
        recv_layout = gen_layer_comm_groups([strategies[lo - 1]], world, rank,
                                            cache)[0][0]
    if cfg.train.deterministic_mode:
        for blk in blocks:
            r = getattr(getattr(blk.inner, "mlp", None), "router", None)
            if r is not None:
                r.deterministic = True
    sm = StageModel(
        blocks=blocks, stage=my_stage, pp_deg=pp, world_size=world, rank=rank,
        plan=plan, embed_comm_group=embed_group, cache=cache,
        recv_layout=recv_layout,
        send_layout=layer_groups[hi - 1] if hi > lo else prev,
        tied_embedding_pair=(emb_block, head_block)
        if (emb_block is not None and head_block is not None) else None,
        # t5 pp boundaries: a cut before layer n_enc carries an
        # encoder-shaped activation; a cut after the bridge additionally
        # carries the full-seq encoder memory (engine cats it onto the
        # boundary rows; uniform dp keeps its batch layout rank-invariant)
        recv_is_encoder=is_encdec and my_stage > 0 and lo - 1 < n_enc,
        recv_carries_memory=is_encdec and my_stage > 0 and lo > n_enc,
        send_carries_memory=is_encdec and my_stage < pp - 1 and hi > n_enc,
    )
    return sm
