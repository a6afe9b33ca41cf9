"""Pipeline execution engine: no-pipeline, GPipe, 1F1B (pipedream-flush).

Reference: galvatron/core/runtime/pipeline/pipeline.py:43-1604.
Grad synchronization is explicit (the ZeRO engine's start/finish_grad_sync
after the last microbatch), replacing FSDP no_sync juggling; the final
microbatch's backward kicks per-block async reductions so DP comm overlaps
the remaining backward compute (grad_reduce.py:48-153 semantics).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..models.builder import StageModel
from . import p2p


def boundary_shape(layout, batch_size: int, seq_len: int, hidden: int):
    """Activation shape at a stage boundary under `layout` (canonical SBH
    sharding: seq / (cp*tsp), batch / dp)."""
    s = layout.strategy
    rows = seq_len // (s.cp * s.tp_sp)
    b_loc = batch_size // s.dp
    return (rows, b_loc, hidden)


def chunk_batch(ctx: Dict, chunks: int, dp: int = 1) -> List[Dict]:
    """Split the global batch context into microbatch contexts
    (reference: pipeline.py:327 chunk_batch + :275 remainder shapes).

    Non-divisible batches are allowed: each microbatch's size is a
    multiple of `dp` (the data-parallel split inside the model), with the
    first `rem` microbatches one dp-row larger.  Per-microbatch p2p shape
    negotiation downstream keys off ctx["batch_size"], so remainder
    microbatches flow through GPipe/1F1B unchanged."""
    B = ctx["batch_size"]
    assert B % dp == 0, f"global batch {B} not divisible by dp {dp}"
    rows = B // dp
    chunks = min(chunks, rows)
    base, rem = divmod(rows, chunks)
    sizes = [(base + (1 if m < rem else 0)) * dp for m in range(chunks)]
    out, lo = [], 0
    for b_mb in sizes:
        sub = dict(ctx)
        sub["input_ids"] = ctx["input_ids"][lo:lo + b_mb]
        sub["labels"] = ctx["labels"][lo:lo + b_mb]
        if "enc_input_ids" in ctx:
            sub["enc_input_ids"] = ctx["enc_input_ids"][lo:lo + b_mb]
        if "loss_mask" in ctx:
            sub["loss_mask"] = ctx["loss_mask"][lo:lo + b_mb]
        sub["batch_size"] = b_mb
        out.append(sub)
        lo += b_mb
    return out


@dataclass
class StepStats:
    loss_sum: float = 0.0
    token_count: float = 0.0

    @property
    def loss(self) -> float:
        return self.loss_sum / max(self.token_count, 1.0)


class PipelineEngine:
    """Drives one optimizer-step's forward+backward over the stage blocks."""

    def __init__(self, stage_model: StageModel, hidden_size: int,
                 pipeline_type: str = "pipedream_flush",
                 overlap_grad_reduce: bool = True,
                 act_dtype: torch.dtype = torch.bfloat16):
        self.sm = stage_model
        self.hidden = hidden_size
        self.pipeline_type = pipeline_type
        self.overlap = overlap_grad_reduce
        self.act_dtype = act_dtype
        self.profiler = None  # RuntimeProfiler: fwd_start/fwd_end brackets
        self.loss_scale = 1.0  # fp16 dynamic loss scaling (LossScaler)
        self._after_fwd_snapped = False
        self._pending_sends = []  # [(reqs, tensor)] in-flight async p2p
        self.device = next(stage_model.parameters()).device \
            if any(True for _ in stage_model.parameters()) else torch.device("cpu")
        world = stage_model.world_size
        G = world // stage_model.pp_deg
        idx = stage_model.rank % G
        self.prev_rank = (stage_model.stage - 1) * G + idx if stage_model.stage > 0 else None
        self.next_rank = (stage_model.stage + 1) * G + idx \
            if stage_model.stage < stage_model.pp_deg - 1 else None

    def _send_async(self, t, dst) -> None:
        self._pending_sends.append(p2p.send_tensor_async(t, dst))

    def _drain_sends(self) -> None:
        for reqs, _t in self._pending_sends:
            for r in reqs:
                r.wait()
        self._pending_sends = []

    # ------------------------------------------------------------------ steps

    def _forward_chunk(self, ctx: Dict, recv_act: Optional[torch.Tensor]):
        if self.profiler is not None:
            self.profiler.fwd_start()
        h = recv_act
        for blk in self.sm.blocks:
            h = blk(h, ctx)
        if self.profiler is not None:
            self.profiler.fwd_end()
        return h

    def _loss_of(self, per_token: torch.Tensor, ctx: Dict, chunks: int):
        denom = ctx.get("loss_denom") or \
            float(ctx["seq_len"]) * ctx["global_batch"]
        loss = per_token.float().sum() / denom
        if self.loss_scale != 1.0:
            loss = loss * self.loss_scale
        return loss

    def _stat_update(self, stats: StepStats, per_token: torch.Tensor,
                     ctx: Dict = None) -> None:
        with torch.no_grad():
            stats.loss_sum += per_token.float().sum().item()
            n = (ctx or {}).pop("_loss_count_local", None)
            stats.token_count += n if n is not None else per_token.numel()

    def _set_auto_sync(self, flag: bool) -> None:
        for blk in self.sm.blocks:
            for f in (blk.flat, getattr(blk, "flat_expert", None)):
                if f is not None:
                    f.auto_sync = flag and self.overlap

    def _finalize_grads(self) -> None:
        for blk in self.sm.blocks:
            blk.finalize_backward()
        self._sync_tied_embeddings_pre()
        for blk in self.sm.blocks:
            for f in (blk.flat, getattr(blk, "flat_expert", None)):
                if f is not None:
                    f.start_grad_sync()
        for blk in self.sm.blocks:
            for f in (blk.flat, getattr(blk, "flat_expert", None)):
                if f is not None:
                    f.finish_grad_sync()

    def _sync_tied_embeddings_pre(self) -> None:
        """pp>1 tied embedding/lm-head: sum the tied segments' raw grads over
        the embedding group BEFORE sdp reduction (sums commute)
        (reference: grad_reduce.py:69-130)."""
        sm = self.sm
        if sm.embed_comm_group is None or sm.embed_comm_group.size == 1:
            return
        tied_param = None
        for blk in self.sm.blocks:
            margs = getattr(blk.inner, "margs", None)
            if margs is None or not getattr(margs, "tie_word_embeddings", False) \
                    or getattr(margs, "untie_embeddings_and_output_weights", True):
                continue
            if blk.kind == "embedding":
                tied_param = blk.inner.word_embeddings.weight
            elif blk.kind == "lm_head":
                tied_param = blk.inner.lm_head.weight
        if tied_param is None:
            return
        owner = getattr(tied_param, "_galvatron_owner", None)
        if owner is None or owner.flat_grad is None:
            return
        for seg in owner.segments:
            if seg.param is tied_param:
                sl = owner.flat_grad[seg.offset:seg.offset + seg.numel]
                dist.all_reduce(sl, group=sm.embed_comm_group.group)

    # -------------------------------------------------------------- schedules

    def forward_backward(self, ctx: Dict, chunks: int) -> StepStats:
        ctx = dict(ctx)
        ctx["global_batch"] = ctx["batch_size"]
        # every layer's dp must divide each microbatch (degrees are powers
        # of two, so the max is the lcm)
        dp = max((blk.groups.strategy.dp for blk in self.sm.blocks
                  if getattr(blk, "groups", None) is not None), default=1)
        mb = chunk_batch(ctx, chunks, dp)
        for m in mb:
            m["global_batch"] = ctx["global_batch"]
        stats = StepStats()
        from ..moe.router import MoEAuxLossAutoScaler
        MoEAuxLossAutoScaler.main_loss_backward_scale = \
            self.loss_scale / max(len(mb), 1)
        self._set_auto_sync(False)
        self._after_fwd_snapped = False
        if self.sm.pp_deg == 1:
            self._no_pipeline(mb, stats)
        elif self.pipeline_type == "gpipe":
            self._gpipe(mb, stats)
        else:
            self._pipedream_flush(mb, stats)
        self._drain_sends()
        self._finalize_grads()
        return stats

    def evaluate(self, ctx: Dict, chunks: int) -> StepStats:
        """Forward-only pass (validation): microbatched, p2p forwards
        between stages, no autograd graph.  Loss statistics accumulate on
        the last stage exactly as in training."""
        ctx = dict(ctx)
        ctx["global_batch"] = ctx["batch_size"]
        dp = max((blk.groups.strategy.dp for blk in self.sm.blocks
                  if getattr(blk, "groups", None) is not None), default=1)
        mb = chunk_batch(ctx, chunks, dp)
        for m in mb:
            m["global_batch"] = ctx["global_batch"]
        stats = StepStats()
        with torch.no_grad():
            for m in mb:
                recv = None
                if not self.sm.is_first:
                    recv = p2p.recv_tensor(self._recv_shape(m),
                                           self.act_dtype, self.prev_rank,
                                           self.device)
                _inp, out = self._fwd_step(m, stats, recv)
                if not self.sm.is_last:
                    self._send_async(out, self.next_rank)
            self._drain_sends()
        return stats

    def _no_pipeline(self, mb: List[Dict], stats: StepStats) -> None:
        """reference: pipeline.py:306 no_pipeline_forward_backward."""
        n = len(mb)
        for m, ctx in enumerate(mb):
            if m == n - 1:
                self._set_auto_sync(True)
            per_token = self._forward_chunk(ctx, None)
            self._stat_update(stats, per_token, ctx)
            loss = self._loss_of(per_token, ctx, n)
            self._snap_after_fwd()
            loss.backward()

    # -- pp helpers

    def _recv_shape(self, ctx: Dict):
        """Boundary rows at this stage's input.  t5: a cut inside the
        encoder is encoder-seq shaped; a cut past the bridge appends the
        full-seq encoder memory (rows concatenated by the sender)."""
        seq = ctx["enc_input_ids"].shape[1] if self.sm.recv_is_encoder \
            else ctx["seq_len"]
        rows, b_loc, h = boundary_shape(self.sm.recv_layout,
                                        ctx["batch_size"], seq, self.hidden)
        if self.sm.recv_carries_memory:
            rows += ctx["enc_input_ids"].shape[1]
        return (rows, b_loc, h)

    def _fwd_step(self, ctx: Dict, stats: StepStats, recv_act):
        with torch.profiler.record_function("galvatron::fwd_chunk"):
            return self._fwd_step_inner(ctx, stats, recv_act)

    def _fwd_step_inner(self, ctx: Dict, stats: StepStats, recv_act):
        feed = None
        if recv_act is not None:
            recv_act = recv_act.detach().requires_grad_(True)
            feed = recv_act
            if self.sm.recv_carries_memory:
                s_enc = ctx["enc_input_ids"].shape[1]
                ctx["encoder_memory"] = recv_act[-s_enc:]
                feed = recv_act[:-s_enc]
        out = self._forward_chunk(ctx, feed)
        if self.sm.is_last:
            self._stat_update(stats, out, ctx)
        elif self.sm.send_carries_memory:
            # ride the memory along the boundary; autograd routes the
            # receiver's grad back into both the stage output and the
            # memory's producers (bridge or upstream boundary)
            out = torch.cat([out, ctx["encoder_memory"]], dim=0)
        return recv_act, out

    def _snap_after_fwd(self) -> None:
        """Record the After-Fwd memory point once per step, just before the
        first backward — the moment stored activations peak (the memory
        profile's activation_mb = After-Fwd minus Before-Fwd)."""
        if self.profiler is not None and not self._after_fwd_snapped:
            self.profiler.profile_memory("After-Fwd")
            self._after_fwd_snapped = True

    def _bwd_step(self, inp, out, grad_out, ctx: Dict, chunks: int):
        self._snap_after_fwd()
        with torch.profiler.record_function("galvatron::bwd_chunk"):
            return self._bwd_step_inner(inp, out, grad_out, ctx, chunks)

    def _bwd_step_inner(self, inp, out, grad_out, ctx: Dict, chunks: int):
        if self.sm.is_last:
            loss = self._loss_of(out, ctx, chunks)
            loss.backward()
        else:
            torch.autograd.backward(out, grad_tensors=grad_out)
        return inp.grad if inp is not None else None

    def _gpipe(self, mb: List[Dict], stats: StepStats) -> None:
        """reference: pipeline.py:729 gpipe_forward + :836 gpipe_backward."""
        n = len(mb)
        saved: List[Tuple] = []
        for ctx in mb:
            recv = None
            if not self.sm.is_first:
                recv = p2p.recv_tensor(self._recv_shape(ctx), self.act_dtype,
                                       self.prev_rank, self.device)
            inp, out = self._fwd_step(ctx, stats, recv)
            if not self.sm.is_last:
                self._send_async(out, self.next_rank)
            saved.append((inp, out, ctx))
        for m, (inp, out, ctx) in enumerate(saved):
            if m == n - 1:
                self._set_auto_sync(True)
            grad_out = None
            if not self.sm.is_last:
                grad_out = p2p.recv_tensor(tuple(out.shape), out.dtype,
                                           self.next_rank, self.device)
            din = self._bwd_step(inp, out, grad_out, ctx, n)
            if not self.sm.is_first and din is not None:
                self._send_async(din, self.prev_rank)
            saved[m] = None

    def _pipedream_flush(self, mb: List[Dict], stats: StepStats) -> None:
        """1F1B (reference: pipeline.py:386 pipedream_flush_forward_backward)."""
        n = len(mb)
        pp, stage = self.sm.pp_deg, self.sm.stage
        num_warmup = min(pp - stage - 1, n)
        num_steady = n - num_warmup
        saved: List[Tuple] = []
        fwd_i = 0
        bwd_i = 0
        # ---- warmup forwards
        for _ in range(num_warmup):
            ctx = mb[fwd_i]
            recv = None
            if not self.sm.is_first:
                recv = p2p.recv_tensor(self._recv_shape(ctx), self.act_dtype,
                                       self.prev_rank, self.device)
            inp, out = self._fwd_step(ctx, stats, recv)
            if not self.sm.is_last:
                self._send_async(out, self.next_rank)
            saved.append((inp, out, ctx))
            fwd_i += 1
        # ---- steady 1F1B
        for k in range(num_steady):
            ctx = mb[fwd_i]
            recv = None
            if not self.sm.is_first:
                recv = p2p.recv_tensor(self._recv_shape(ctx), self.act_dtype,
                                       self.prev_rank, self.device)
            inp, out = self._fwd_step(ctx, stats, recv)
            saved.append((inp, out, ctx))
            fwd_i += 1
            # send fwd + recv bwd (fused where both present)
            grad_out = None
            b_inp, b_out, b_ctx = saved[bwd_i]
            if not self.sm.is_last:
                grad_out, sreqs, stens = p2p.send_recv_async(
                    out, self.next_rank, tuple(b_out.shape), b_out.dtype,
                    self.next_rank, self.device)
                if sreqs:
                    self._pending_sends.append((sreqs, stens))
            if bwd_i == n - 1:
                self._set_auto_sync(True)
            din = self._bwd_step(b_inp, b_out, grad_out, b_ctx, n)
            saved[bwd_i] = None
            bwd_i += 1
            if not self.sm.is_first and din is not None:
                self._send_async(din, self.prev_rank)
        # ---- cooldown backwards
        while bwd_i < n:
            b_inp, b_out, b_ctx = saved[bwd_i]
            grad_out = None
            if not self.sm.is_last:
                grad_out = p2p.recv_tensor(tuple(b_out.shape), b_out.dtype,
                                           self.next_rank, self.device)
            if bwd_i == n - 1:
                self._set_auto_sync(True)
            din = self._bwd_step(b_inp, b_out, grad_out, b_ctx, n)
            saved[bwd_i] = None
            bwd_i += 1
            if not self.sm.is_first and din is not None:
                self._send_async(din, self.prev_rank)
