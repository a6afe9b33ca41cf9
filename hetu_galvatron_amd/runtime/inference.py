"""Single-GPU serving path: KV-cache autoregressive generation.

Reference role: the optional flash-decode inference dependency
(nvidia_chunked_flash_attn, galvatron attention.py:398-514) — the
reference treats decode as an optional attention backend; here it is a
first-class engine over the same hybrid-parallel module weights.

MI355X design: decode attention is one memory-bound HIP kernel
(`decode_attn` in ops/csrc/elementwise.hip) streaming the bf16 KV cache
at HBM rate; prefill reuses the MFMA flash kernel while writing the
cache.  288 GB HBM3E comfortably holds an 8B model + tens of GB of KV
cache on ONE GPU, so the primary serving topology is world=1 (tp/pp/cp=1,
any dp_type — at world 1 every flat-param mode keeps full params
resident).  GalvatronTPGenerator adds megatron-TP decode across GPUs
(world == tp) for models beyond one GPU's memory.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..ops import decode_attention, flash_attention_fwd_only

__all__ = ["KVCache", "GalvatronGenerator", "GalvatronTPGenerator"]


class KVCache:
    """Per-layer bf16 K/V ring of shape [b, max_seq, hkv, d]."""

    def __init__(self, n_layers: int, b: int, max_seq: int, hkv: int, d: int,
                 device, dtype=torch.bfloat16):
        self.k = [torch.empty(b, max_seq, hkv, d, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.v = [torch.empty(b, max_seq, hkv, d, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.max_seq = max_seq
        self.cur_len = 0

    def write(self, layer: int, k: torch.Tensor, v: torch.Tensor) -> None:
        """k, v: [b, s_new, hkv, d] for positions [cur_len, cur_len + s_new)."""
        s = k.shape[1]
        self.k[layer][:, self.cur_len:self.cur_len + s] = k
        self.v[layer][:, self.cur_len:self.cur_len + s] = v

    def advance(self, s: int) -> None:
        self.cur_len += s
        assert self.cur_len <= self.max_seq, "KV cache overflow"


class GalvatronGenerator:
    """Greedy / temperature sampling over a built GalvatronModel.

    Drives the trained module weights directly (embedding -> decoder
    layers -> final norm -> lm head), bypassing the training engine's
    loss head.  Requires a world-size-1 model (serving topology v1).
    """

    def __init__(self, model, max_batch: int = 1, max_seq: int = 4096):
        sm = model.stage_model
        assert sm.world_size == 1 and sm.pp_deg == 1, \
            "GalvatronGenerator v1 serves on one GPU (world=1)"
        self.cfg = model.cfg
        m = self.cfg.model
        assert m.position_embedding_type == "rope", \
            "generator v1 supports RoPE models (llama/mistral/qwen family)"
        self.margs = m
        self.blocks = sm.blocks
        kinds = [b.kind for b in self.blocks]
        assert kinds[0] == "embedding" and kinds[-1] == "lm_head"
        self.layers = [b.inner for b in self.blocks
                       if b.kind in ("decoder", "encoder")]
        self.embedding = self.blocks[0].inner
        self.final_norm = next(b.inner for b in self.blocks
                               if b.kind == "final_norm")
        self.lm_head = self.blocks[-1].inner
        self.max_batch = max_batch
        self.max_seq = max_seq
        # mistral sliding window: windowed decode + prefill are native
        # (graph capture still needs max_seq <= window, asserted there)
        self.window = getattr(m, "sliding_window", None)
        self.scale = 1.0 / (m.head_dim ** 0.5)
        self._dev = next(self.embedding.parameters()).device
        self._graphs: dict = {}   # batch size -> captured decode step
        self._kv: dict = {}       # batch size -> persistent KVCache

    # -- one decoder layer, cache-aware ------------------------------------
    @staticmethod
    def _split_qkv(attn, qkv: torch.Tensor):
        """Interleaved-group [q*(hq/hkv), k, v] layout (attention.py)."""
        s, b = qkv.shape[0], qkv.shape[1]
        qkv = qkv.view(s, b, attn.num_groups_local, attn.q_per_group + 2,
                       attn.head_dim)
        q = qkv[:, :, :, : attn.q_per_group].reshape(s, b, -1, attn.head_dim)
        k = qkv[:, :, :, attn.q_per_group].reshape(s, b, -1, attn.head_dim)
        v = qkv[:, :, :, attn.q_per_group + 1].reshape(s, b, -1, attn.head_dim)
        return q, k, v

    def _layer_step(self, li: int, hidden: torch.Tensor, cache: KVCache,
                    pos: int) -> torch.Tensor:
        """hidden: [s_new, b, h] for absolute positions [pos, pos+s_new)."""
        layer = self.layers[li]
        attn = layer.attention
        residual = hidden
        x = layer.input_norm(hidden)
        qkv = attn.linear_qkv(x)
        q, k, v = self._split_qkv(attn, qkv)
        if getattr(attn, "q_layernorm", None) is not None:
            q = attn.q_layernorm(q)
            k = attn.k_layernorm(k)
        if layer.rotary is not None:
            cos, sin = layer.rotary.full_tables(pos + q.shape[0], hidden.device)
            from .transformer.rope import apply_rope_qk
            q, k = apply_rope_qk(q.contiguous(), k.contiguous(),
                                 cos[pos:], sin[pos:],
                                 interleaved=getattr(
                                     attn, "rope_interleaved", False))
        # cache layout [b, s, hkv, d]
        cache.write(li, k.permute(1, 0, 2, 3), v.permute(1, 0, 2, 3))
        s_new = q.shape[0]
        win = getattr(self, "window", None)
        if s_new == 1:
            o = decode_attention(q[0], cache.k[li], cache.v[li],
                                 pos + 1, softmax_scale=self.scale,
                                 window=win)
            o = o.unsqueeze(0)  # [1, b, hq, d]
        elif pos == 0:
            qb = q.permute(1, 0, 2, 3).contiguous()
            kb = k.permute(1, 0, 2, 3).contiguous()
            vb = v.permute(1, 0, 2, 3).contiguous()
            ob, _ = flash_attention_fwd_only(qb, kb, vb, causal=True,
                                             softmax_scale=self.scale,
                                             window=win)
            o = ob.permute(1, 0, 2, 3)
        else:
            # chunked prefill continuation: one cross-length flash call —
            # the new s_new queries attend the whole [0, pos+s_new) cache
            # with bottom-right causal alignment (kernel supports sq<skv)
            qb = q.permute(1, 0, 2, 3).contiguous()
            kb = cache.k[li][:, :pos + s_new].contiguous()
            vb = cache.v[li][:, :pos + s_new].contiguous()
            ob, _ = flash_attention_fwd_only(qb, kb, vb, causal=True,
                                             softmax_scale=self.scale,
                                             window=win)
            o = ob.permute(1, 0, 2, 3)
        o = o.reshape(s_new, o.shape[1], -1)
        x = attn.linear_proj(o)
        hidden = residual + x
        residual = hidden
        x = layer.post_attn_norm(hidden)
        x = layer.mlp(x)
        return residual + x

    @torch.no_grad()
    def _forward_tokens(self, tokens: torch.Tensor, cache: KVCache
                        ) -> torch.Tensor:
        """tokens: [b, s_new] at positions [cache.cur_len, ...); returns
        last-position logits [b, V]."""
        pos = cache.cur_len
        h = self.embedding.word_embeddings(tokens)  # [s_new, b, h]
        for li in range(len(self.layers)):
            h = self._layer_step(li, h, cache, pos)
        cache.advance(tokens.shape[1])
        h = self.final_norm.norm(h[-1:])
        logits = self.lm_head.lm_head(h)  # [1, b, V]
        return logits[0].float()

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_id: Optional[int] = None) -> torch.Tensor:
        """input_ids: [b, s_prompt] on the model device.  Returns
        [b, s_prompt + n_generated] (greedy when temperature == 0)."""
        b, sp = input_ids.shape
        assert b <= self.max_batch and sp + max_new_tokens <= self.max_seq
        m = self.margs
        cache = KVCache(len(self.layers), b, self.max_seq, m.kv_heads,
                        m.head_dim, self._dev,
                        dtype=next(self.embedding.parameters()).dtype)
        for blk in self.blocks:
            if blk.flat is not None:
                blk.flat.gather_params()
        was_training = [l.training for l in self.layers]
        for blk in self.blocks:
            blk.inner.eval()
        try:
            logits = self._forward_tokens(input_ids, cache)
            out: List[torch.Tensor] = [input_ids]
            done = torch.zeros(b, dtype=torch.bool, device=self._dev)
            for _ in range(max_new_tokens):
                nxt = self._sample(logits, temperature, top_k)
                if eos_id is not None:
                    nxt = torch.where(done, torch.full_like(nxt, eos_id), nxt)
                    done |= nxt.eq(eos_id)
                out.append(nxt.unsqueeze(1))
                if eos_id is not None and bool(done.all()):
                    break
                logits = self._forward_tokens(nxt.unsqueeze(1), cache)
        finally:
            for l, t in zip(self.layers, was_training):
                l.train(t)
        return torch.cat(out, dim=1)

    # ------------------------------------------------------------ hipGraph
    @torch.no_grad()
    def generate_graphed(self, input_ids: torch.Tensor,
                         max_new_tokens: int = 32,
                         warmup_steps: int = 3) -> torch.Tensor:
        """Greedy generation with the decode step captured in ONE hipGraph.
        Windowed models: the captured decode attends the full cache, so
        the capture requires max_seq <= window (eager generate() handles
        windowed decode natively)."""
        if self.window is not None:
            assert self.max_seq <= self.window, \
                "generate_graphed: max_seq must fit the sliding window " \
                "(use generate() for windowed decode)"
        return self._generate_graphed_impl(input_ids, max_new_tokens,
                                           warmup_steps)

    def _generate_graphed_impl(self, input_ids: torch.Tensor,
                               max_new_tokens: int = 32,
                               warmup_steps: int = 3) -> torch.Tensor:
        """(implementation; see generate_graphed docstring)

        The step is fully device-driven — position index, KV-cache
        index_copy, decode kernel length (decode_attn_graph reads it from
        device memory), argmax and its feedback into the next step's
        input all live inside the capture — so each new token is a single
        graph replay with zero host work (the eager loop costs ~9 ms/token
        in Python/launch overhead vs ~0.5 ms of kernel time).  Falls back
        to the eager path if capture fails.
        """
        b, sp = input_ids.shape
        assert sp + max_new_tokens <= self.max_seq
        m = self.margs
        dev = self._dev
        # persistent per-batch-size cache: the captured graph holds device
        # pointers into these tensors, so reuse across requests is what
        # makes capture a one-time cost
        cache = self._kv.get(b)
        if cache is None:
            cache = KVCache(len(self.layers), b, self.max_seq, m.kv_heads,
                            m.head_dim, dev,
                            dtype=next(self.embedding.parameters()).dtype)
            self._kv[b] = cache
        cache.cur_len = 0
        for blk in self.blocks:
            if blk.flat is not None:
                blk.flat.gather_params()
        was_training = [l.training for l in self.layers]
        for blk in self.blocks:
            blk.inner.eval()
        try:
            logits = self._forward_tokens(input_ids, cache)  # prefill
            toks = [logits.argmax(-1)]
            # eager head start: real decode steps double as hipBLASLt /
            # allocator warmup for the capture
            n_eager = min(warmup_steps, max_new_tokens - 1)
            for _ in range(n_eager):
                logits = self._forward_tokens(toks[-1].unsqueeze(1), cache)
                toks.append(logits.argmax(-1))
            n_graph = max_new_tokens - 1 - n_eager
            if n_graph > 0:
                out_g = self._run_graph_steps(toks[-1], cache, n_graph)
                toks.extend(out_g)
            return torch.cat([input_ids] +
                             [t.unsqueeze(1) for t in toks[:max_new_tokens]],
                             dim=1)
        finally:
            for l, t in zip(self.layers, was_training):
                l.train(t)

    def _graph_step(self, ids_buf, pos_i64, cur32, cache: KVCache):
        """One decode step on static buffers (capture-safe)."""
        from ..ops._ext import get_ext
        from ..ops import apply_rope
        assert not getattr(self.layers[0].attention, "rope_interleaved",
                           False), \
            "graphed decode uses the fused NEOX RoPE kernel; " \
            "rotary_interleaved models must use generate()"
        ext = get_ext()
        h = self.embedding.word_embeddings(ids_buf)  # [1, b, h]
        for li, layer in enumerate(self.layers):
            attn = layer.attention
            residual = h
            x = layer.input_norm(h)
            qkv = attn.linear_qkv(x)
            q, k, v = self._split_qkv(attn, qkv)
            if getattr(attn, "q_layernorm", None) is not None:
                q = attn.q_layernorm(q)
                k = attn.k_layernorm(k)
            if layer.rotary is not None:
                cos_t, sin_t = layer.rotary.full_tables(self.max_seq,
                                                        h.device)
                cos = cos_t.index_select(0, pos_i64)
                sin = sin_t.index_select(0, pos_i64)
                q = apply_rope(q.contiguous(), cos, sin)
                k = apply_rope(k.contiguous(), cos, sin)
            cache.k[li].index_copy_(1, pos_i64, k.permute(1, 0, 2, 3))
            cache.v[li].index_copy_(1, pos_i64, v.permute(1, 0, 2, 3))
            o = ext.decode_attn_graph(q[0].contiguous(), cache.k[li],
                                      cache.v[li], cur32, self.max_seq,
                                      self.scale)
            o = o.reshape(1, o.shape[0], -1)
            h = residual + attn.linear_proj(o)
            residual = h
            h = residual + layer.mlp(layer.post_attn_norm(h))
        h = self.final_norm.norm(h)
        logits = self.lm_head.lm_head(h)[0].float()
        nxt = logits.argmax(-1)  # [b]
        ids_buf.copy_(nxt.unsqueeze(1))
        pos_i64.add_(1)
        cur32.add_(1)
        return nxt

    def _run_graph_steps(self, first_ids: torch.Tensor, cache: KVCache,
                         n_steps: int):
        """Replay the captured decode step n_steps times.  The capture is
        keyed by batch size and reused across generate_graphed calls (the
        graph holds pointers into THIS generator's persistent KV cache, so
        callers must pass the cache generate_graphed allocated)."""
        dev = self._dev
        b = first_ids.shape[0]
        ent = self._graphs.get(b)
        if ent is not None and ent[4].shape[0] != self.max_seq:
            ent = None  # max_seq changed: stale capture
        if ent is None:
            ids_buf = first_ids.unsqueeze(1).clone()          # [b, 1]
            pos_i64 = torch.tensor([cache.cur_len], dtype=torch.long,
                                   device=dev)
            cur32 = torch.tensor([cache.cur_len + 1], dtype=torch.int32,
                                 device=dev)
            out_buf = torch.empty(self.max_seq, b, dtype=first_ids.dtype,
                                  device=dev)
            step_i = torch.zeros(1, dtype=torch.long, device=dev)
            for layer in self.layers:  # tables must exist before capture
                if layer.rotary is not None:
                    layer.rotary.full_tables(self.max_seq, dev)
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    nxt = self._graph_step(ids_buf, pos_i64, cur32, cache)
                    out_buf.index_copy_(0, step_i, nxt.unsqueeze(0))
                    step_i.add_(1)
            except Exception:
                # graph capture unavailable: eager fallback
                out = []
                ids = first_ids
                for _ in range(n_steps):
                    logits = self._forward_tokens(ids.unsqueeze(1), cache)
                    ids = logits.argmax(-1)
                    out.append(ids)
                return out
            ent = (g, ids_buf, pos_i64, cur32, out_buf, step_i)
            self._graphs[b] = ent
        g, ids_buf, pos_i64, cur32, out_buf, step_i = ent
        ids_buf.copy_(first_ids.unsqueeze(1))
        pos_i64.fill_(cache.cur_len)
        cur32.fill_(cache.cur_len + 1)
        step_i.zero_()
        for _ in range(n_steps):
            g.replay()
        cache.cur_len = int(pos_i64.item())
        return list(out_buf[:n_steps].unbind(0))

    @staticmethod
    def _sample(logits: torch.Tensor, temperature: float, top_k: int
                ) -> torch.Tensor:
        if temperature <= 0.0:
            return logits.argmax(-1)
        logits = logits / temperature
        if top_k > 0:
            kth = logits.topk(top_k, dim=-1).values[..., -1, None]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        return torch.multinomial(probs, 1).squeeze(-1)


class GalvatronTPGenerator:
    """Megatron-TP decode across GPUs (serving topology v2 -> now v1.5):
    world == tp, dp/pp/cp = 1, non-ulysses.

    Runs the SHARDED module weights directly with replicated single-token
    activations — column linears produce local heads / columns, row
    linears produce partials reduced over tp, the vocab-sharded lm head
    is gathered on the last dim before argmax — so decode needs exactly
    TWO all-reduces + one head gather per token per layer-stack pass
    (the same collective count as a Megatron training step's attention +
    MLP row projections).  Greedy only; every rank computes identical
    tokens (collectives keep them in lockstep).
    """

    def __init__(self, model, max_batch: int = 1, max_seq: int = 4096):
        import torch.distributed as dist
        sm = model.stage_model
        self.world = sm.world_size
        assert sm.pp_deg == 1, "tp decode: pp=1"
        self.cfg = model.cfg
        m = self.cfg.model
        assert m.position_embedding_type == "rope"
        self.margs = m
        assert m.hidden_act in ("silu", "swiglu"), \
            "tp decode implements the silu/swiglu MLP only"
        self.blocks = sm.blocks
        self.layers = [b.inner for b in self.blocks if b.kind == "decoder"]
        self.embedding = self.blocks[0].inner
        self.final_norm = next(b.inner for b in self.blocks
                               if b.kind == "final_norm")
        self.lm_head = self.blocks[-1].inner
        g0 = self.blocks[1].groups
        s0 = g0.strategy
        assert not s0.use_ulysses and s0.cp == 1 and s0.dp == 1 and \
            s0.tp == self.world, \
            "tp decode expects a pure megatron-tp plan (tp == world)"
        self.tp_group = g0.tp_group.group
        self.max_batch = max_batch
        self.max_seq = max_seq
        # windowed decode is native via decode_attention(window=) below
        self.window = getattr(m, "sliding_window", None)
        self.scale = 1.0 / (m.head_dim ** 0.5)
        self._dev = next(self.embedding.parameters()).device

    def _embed(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [b, s] -> replicated [s, b, h] (vocab-sharded table)."""
        import torch.distributed as dist
        import torch.nn.functional as F
        we = self.embedding.word_embeddings
        mask = (ids < we.vocab_start_index) | (ids >= we.vocab_end_index)
        local = (ids - we.vocab_start_index).masked_fill(mask, 0)
        h = F.embedding(local, we.weight).masked_fill(mask.unsqueeze(-1), 0.0)
        dist.all_reduce(h, group=self.tp_group)
        return h.transpose(0, 1).contiguous()  # [s, b, h]

    def _layer_step(self, li: int, hidden, cache: KVCache, pos: int):
        """hidden [s,b,h] replicated; local-head attention + tp-reduced
        projections."""
        import torch.distributed as dist
        import torch.nn.functional as F
        from ..ops import swiglu
        layer = self.layers[li]
        attn = layer.attention
        residual = hidden
        x = layer.input_norm(hidden)
        qkv = F.linear(x, attn.linear_qkv.weight,
                       getattr(attn.linear_qkv, "bias", None))
        q, k, v = GalvatronGenerator._split_qkv(attn, qkv)
        if getattr(attn, "q_layernorm", None) is not None:
            q = attn.q_layernorm(q)
            k = attn.k_layernorm(k)
        if layer.rotary is not None:
            cos, sin = layer.rotary.full_tables(pos + q.shape[0], x.device)
            from .transformer.rope import apply_rope_qk
            q, k = apply_rope_qk(q.contiguous(), k.contiguous(),
                                 cos[pos:], sin[pos:],
                                 interleaved=getattr(
                                     attn, "rope_interleaved", False))
        cache.write(li, k.permute(1, 0, 2, 3), v.permute(1, 0, 2, 3))
        s_new = q.shape[0]
        if s_new == 1:
            o = decode_attention(q[0], cache.k[li], cache.v[li], pos + 1,
                                 softmax_scale=self.scale,
                                 window=self.window).unsqueeze(0)
        else:
            qb = q.permute(1, 0, 2, 3).contiguous()
            kb = cache.k[li][:, :pos + s_new].contiguous()
            vb = cache.v[li][:, :pos + s_new].contiguous()
            ob, _ = flash_attention_fwd_only(qb, kb, vb, causal=True,
                                             softmax_scale=self.scale,
                                             window=self.window)
            o = ob.permute(1, 0, 2, 3)
        o = o.reshape(s_new, o.shape[1], -1)
        part = F.linear(o, attn.linear_proj.weight)  # row-parallel partial
        dist.all_reduce(part, group=self.tp_group)
        if attn.linear_proj.bias is not None:
            part = part + attn.linear_proj.bias
        hidden = residual + part
        residual = hidden
        x = layer.post_attn_norm(hidden)
        h1 = F.linear(x, layer.mlp.fc1.weight,
                      getattr(layer.mlp.fc1, "bias", None))
        h1 = swiglu(h1)
        h2 = F.linear(h1, layer.mlp.fc2.weight)
        dist.all_reduce(h2, group=self.tp_group)
        if layer.mlp.fc2.bias is not None:
            h2 = h2 + layer.mlp.fc2.bias
        return residual + h2

    @torch.no_grad()
    def _forward_tokens(self, tokens, cache: KVCache):
        import torch.distributed as dist
        pos = cache.cur_len
        h = self._embed(tokens)
        for li in range(len(self.layers)):
            h = self._layer_step(li, h, cache, pos)
        cache.advance(tokens.shape[1])
        h = self.final_norm.norm(h[-1:])
        import torch.nn.functional as F
        local = F.linear(h, self.lm_head.lm_head.weight)  # [1,b,V/t]
        parts = [torch.empty_like(local) for _ in range(self.world)]
        dist.all_gather(parts, local.contiguous(), group=self.tp_group)
        return torch.cat(parts, dim=-1)[0].float()  # [b, V]

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, seed: int = 0) -> torch.Tensor:
        """Greedy, or temperature sampling with a seed-shared generator —
        every tp rank draws the identical sample from the identical
        gathered logits, keeping ranks in lockstep."""
        b, sp = input_ids.shape
        assert sp + max_new_tokens <= self.max_seq
        m = self.margs
        tpd = self.world
        cache = KVCache(len(self.layers), b, self.max_seq,
                        max(m.kv_heads // tpd, 1), m.head_dim, self._dev,
                        dtype=next(self.embedding.parameters()).dtype)
        for blk in self.blocks:
            if blk.flat is not None:
                blk.flat.gather_params()
        for blk in self.blocks:
            blk.inner.eval()
        gen = None
        if temperature > 0:
            gen = torch.Generator(device="cpu").manual_seed(seed)
        logits = self._forward_tokens(input_ids, cache)
        out = [input_ids]
        for _ in range(max_new_tokens):
            if temperature > 0:
                probs = torch.softmax(logits.cpu() / temperature, dim=-1)
                nxt = torch.multinomial(probs, 1, generator=gen)                     .squeeze(-1).to(logits.device)
            else:
                nxt = logits.argmax(-1)
            out.append(nxt.unsqueeze(1))
            if len(out) - 1 < max_new_tokens:
                logits = self._forward_tokens(nxt.unsqueeze(1), cache)
        return torch.cat(out, dim=1)
