"""Continuous-batching serving engine over the KV-cache generator.

Requests join and leave between decode steps (vLLM-style scheduling
semantics, reference has no serving path): each slot owns a contiguous
row of the persistent KV cache with its own length, the per-step linears
(qkv, proj, mlp, head) are batched across every active slot, and
attention runs per slot on its cache row ([slot:slot+1] is a contiguous
view — no copies).

v1 performance note: per-slot attention means n_active kernel launches
per layer; the batched-attention upgrade is a per-row cur_len variant of
the decode kernel (roadmap).  RoPE at mixed positions is applied with a
pure-torch broadcast (per-slot table rows), not the shared-table kernel.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..ops import decode_attention, flash_attention_fwd_only
from .inference import GalvatronGenerator, KVCache

__all__ = ["ContinuousBatchingEngine"]


def _rope_rows(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               interleaved: bool = False):
    """x: [1, b, h, d]; cos/sin: [b, d/2] (one table row per slot)."""
    d2 = x.shape[-1] // 2
    c = cos[None, :, None, :].float()
    s = sin[None, :, None, :].float()
    if interleaved:
        x1, x2 = x[..., 0::2].float(), x[..., 1::2].float()
        y = torch.empty_like(x, dtype=torch.float32)
        y[..., 0::2] = x1 * c - x2 * s
        y[..., 1::2] = x2 * c + x1 * s
        return y.to(x.dtype)
    x1, x2 = x[..., :d2].float(), x[..., d2:].float()
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1).to(x.dtype)


class ContinuousBatchingEngine:
    """Greedy decode over a slot pool (world=1 topology)."""

    def __init__(self, model, max_slots: int = 8, max_seq: int = 4096):
        self.gen = GalvatronGenerator(model, max_batch=max_slots,
                                      max_seq=max_seq)
        m = model.cfg.model
        self.margs = m
        self.max_slots = max_slots
        self.max_seq = max_seq
        dtype = next(self.gen.embedding.parameters()).dtype
        self.cache = KVCache(len(self.gen.layers), max_slots, max_seq,
                             m.kv_heads, m.head_dim, self.gen._dev,
                             dtype=dtype)
        self.lengths = [0] * max_slots          # attended tokens per slot
        self.budgets: Dict[int, int] = {}       # slot -> tokens remaining
        self.last_tok: Dict[int, int] = {}      # slot -> pending input token
        self.free: List[int] = list(range(max_slots))
        self.slot_of: Dict[int, int] = {}       # request id -> slot
        self.rid_of: Dict[int, int] = {}        # slot -> request id
        self.outputs: Dict[int, List[int]] = {} # request id -> tokens
        self.samplers: Dict[int, tuple] = {}    # slot -> (temp, generator)
        self.eos: Dict[int, int] = {}           # slot -> eos id (or None)
        self._next_rid = 0
        for blk in self.gen.blocks:
            if blk.flat is not None:
                blk.flat.gather_params()
            blk.inner.eval()

    # -- request lifecycle --------------------------------------------------
    def add_request(self, prompt_ids: torch.Tensor, max_new_tokens: int,
                    temperature: float = 0.0, seed: int = 0,
                    eos_id: int = None) -> int:
        """prompt_ids: [s]; prefills a slot, producing the request's
        FIRST generated token immediately.  Returns a request id whose
        tokens accumulate in `self.outputs[rid]` (kept after release).
        temperature > 0 samples with a per-request seeded generator."""
        assert self.free, "no free slots"
        assert prompt_ids.dim() == 1 and max_new_tokens >= 1
        slot = self.free.pop()
        rid = self._next_rid
        self._next_rid += 1
        sp = prompt_ids.shape[0]
        assert sp + max_new_tokens <= self.max_seq
        # single-slot prefill on the slot's contiguous cache row
        view = _SlotCache(self.cache, slot)
        logits = self._prefill(prompt_ids.unsqueeze(0), view)
        if temperature > 0:
            g = torch.Generator(device="cpu").manual_seed(seed)
            self.samplers[slot] = (temperature, g)
            first = self._sample(logits[0], slot)
        else:
            self.samplers.pop(slot, None)
            first = int(logits.argmax(-1))
        self.outputs[rid] = [first]
        self.slot_of[rid] = slot
        self.rid_of[slot] = rid
        self.lengths[slot] = sp
        self.last_tok[slot] = first
        self.eos[slot] = eos_id
        self.budgets[slot] = max_new_tokens - 1
        if self.budgets[slot] == 0 or (eos_id is not None
                                       and first == eos_id):
            self.release(rid)
        return rid

    def _prefill(self, ids: torch.Tensor, view: "_SlotCache"):
        gen = self.gen
        h = gen.embedding.word_embeddings(ids)
        for li, layer in enumerate(gen.layers):
            attn = layer.attention
            residual = h
            x = layer.input_norm(h)
            q, k, v = gen._split_qkv(attn, attn.linear_qkv(x))
            if getattr(attn, "q_layernorm", None) is not None:
                q = attn.q_layernorm(q)
                k = attn.k_layernorm(k)
            if layer.rotary is not None:
                cos, sin = layer.rotary.full_tables(q.shape[0], h.device)
                from .transformer.rope import apply_rope_qk
                q, k = apply_rope_qk(q.contiguous(), k.contiguous(),
                                     cos, sin,
                                     interleaved=getattr(
                                         attn, "rope_interleaved", False))
            view.write(li, k.permute(1, 0, 2, 3), v.permute(1, 0, 2, 3), 0)
            qb = q.permute(1, 0, 2, 3).contiguous()
            kb = k.permute(1, 0, 2, 3).contiguous()
            vb = v.permute(1, 0, 2, 3).contiguous()
            ob, _ = flash_attention_fwd_only(
                qb, kb, vb, causal=True, softmax_scale=gen.scale,
                window=getattr(gen, "window", None))
            o = ob.permute(1, 0, 2, 3).reshape(q.shape[0], 1, -1)
            h = residual + attn.linear_proj(o)
            residual = h
            h = residual + layer.mlp(layer.post_attn_norm(h))
        h = gen.final_norm.norm(h[-1:])
        return gen.lm_head.lm_head(h)[0].float()

    @torch.no_grad()
    def step(self) -> Dict[int, int]:
        """One decode step for every active slot.  Returns {rid: token};
        requests whose budget hits zero are released after their token
        (their outputs stay readable in `self.outputs`)."""
        active = sorted(self.budgets)
        if not active:
            return {}
        gen = self.gen
        dev = gen._dev
        ids = torch.tensor([[self.last_tok[s] for s in active]],
                           device=dev)                      # [1, n]
        pos = torch.tensor([self.lengths[s] for s in active], device=dev)
        h = gen.embedding.word_embeddings(ids.transpose(0, 1))  # [1, n, hd]
        for li, layer in enumerate(gen.layers):
            attn = layer.attention
            residual = h
            x = layer.input_norm(h)
            q, k, v = gen._split_qkv(attn, attn.linear_qkv(x))
            if getattr(attn, "q_layernorm", None) is not None:
                q = attn.q_layernorm(q)
                k = attn.k_layernorm(k)
            if layer.rotary is not None:
                cos_t, sin_t = layer.rotary.full_tables(self.max_seq, dev)
                il = getattr(attn, "rope_interleaved", False)
                q = _rope_rows(q, cos_t[pos], sin_t[pos], il)
                k = _rope_rows(k, cos_t[pos], sin_t[pos], il)
            outs = []
            for i, s in enumerate(active):
                L = self.lengths[s]
                self.cache.k[li][s, L] = k[0, i]
                self.cache.v[li][s, L] = v[0, i]
                o = decode_attention(q[:, i], self.cache.k[li][s:s + 1],
                                     self.cache.v[li][s:s + 1], L + 1,
                                     softmax_scale=gen.scale,
                                     window=getattr(gen, "window", None))
                outs.append(o)
            o = torch.cat(outs, 0).unsqueeze(0).reshape(1, len(active), -1)
            h = residual + attn.linear_proj(o)
            residual = h
            h = residual + layer.mlp(layer.post_attn_norm(h))
        h = gen.final_norm.norm(h)
        logits = gen.lm_head.lm_head(h)[0].float()          # [n, V]
        greedy = logits.argmax(-1)
        out: Dict[int, int] = {}
        for i, s in enumerate(active):
            t = self._sample(logits[i], s) if s in self.samplers \
                else int(greedy[i])
            rid = self.rid_of[s]
            out[rid] = t
            self.outputs[rid].append(t)
            self.lengths[s] += 1
            self.last_tok[s] = t
            self.budgets[s] -= 1
            if self.budgets[s] == 0 or (self.eos.get(s) is not None
                                        and t == self.eos[s]):
                self.release(rid)
        return out

    def _sample(self, logits_row: torch.Tensor, slot: int) -> int:
        temp, g = self.samplers[slot]
        probs = torch.softmax(logits_row.float().cpu() / temp, dim=-1)
        return int(torch.multinomial(probs, 1, generator=g))

    def release(self, rid: int) -> None:
        slot = self.slot_of.pop(rid, None)
        if slot is None:
            return  # idempotent: already released
        self.samplers.pop(slot, None)
        self.eos.pop(slot, None)
        self.rid_of.pop(slot, None)
        self.budgets.pop(slot, None)
        self.last_tok.pop(slot, None)
        self.lengths[slot] = 0
        self.free.append(slot)

    def collect(self, rid: int) -> List[int]:
        """Return-and-delete a request's output tokens: the long-running
        server path, so `outputs` does not grow without bound."""
        return self.outputs.pop(rid, [])

    @property
    def n_active(self) -> int:
        return len(self.budgets)


class _SlotCache:
    """KVCache adapter exposing one slot's contiguous row."""

    def __init__(self, cache: KVCache, slot: int):
        self.cache = cache
        self.slot = slot

    def write(self, layer: int, k, v, pos: int) -> None:
        s = k.shape[1]
        self.cache.k[layer][self.slot, pos:pos + s] = k[0]
        self.cache.v[layer][self.slot, pos:pos + s] = v[0]
