"""Rotary position embeddings with CP/SP-aware frequency slicing.

Reference: galvatron/core/runtime/transformer/rotary_pos_embedding.py:34-266
and rope_utils.py:86-269.  cos/sin tables are HOST-precomputed (CDNA4
elementwise guideline: no on-device trig in the hot path) and sliced per
rank: Ulysses-SP ranks own a contiguous sequence slice; zigzag-CP ranks own
chunks (i, 2cp-1-i) of 2cp chunks (rotary_pos_embedding.py:34
get_pos_emb_on_this_cp_sp_rank_galvatron).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ...ops import apply_rope
from ...ops.reference_ops import rope_freqs


class RotaryEmbedding(torch.nn.Module):
    """Caches cos/sin [seq, d/2] fp32 tables; returns per-rank slices."""

    def __init__(self, head_dim: int, theta: float = 10000.0,
                 scaling: Optional[float] = None, max_seq: int = 8192):
        super().__init__()
        self.head_dim = head_dim
        self.theta = theta
        self.scaling = scaling
        self._cached: Tuple[int, Optional[torch.Tensor], Optional[torch.Tensor]] = (0, None, None)

    def full_tables(self, seq_len: int, device) -> Tuple[torch.Tensor, torch.Tensor]:
        cached_len, cos, sin = self._cached
        if cos is None or cached_len < seq_len or cos.device != device:
            cos, sin = rope_freqs(seq_len, self.head_dim, self.theta,
                                  device=device, interp=self.scaling or 1.0)
            self._cached = (seq_len, cos, sin)
        return cos[:seq_len], sin[:seq_len]

    def get_for_rank(self, seq_len_global: int, device,
                     sp_rank: int = 0, sp_size: int = 1,
                     cp_rank: int = 0, cp_size: int = 1
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Slice of the tables covering this rank's local sequence shard.

        Layout convention (shared with the dataloader's get_batch slicing):
        the sequence is FIRST zigzag-split over cp ranks (rank's pair of
        chunks), THEN that cp-local sequence is split into sp_size contiguous
        slices (Ulysses / Megatron-SP) — so a Ulysses all-gather reassembles
        exactly the cp rank's zigzag pair.
        """
        cos, sin = self.full_tables(seq_len_global, device)
        if cp_size > 1:
            cos = zigzag_slice(cos, cp_rank, cp_size)
            sin = zigzag_slice(sin, cp_rank, cp_size)
        if sp_size > 1:
            sl = cos.shape[0] // sp_size
            cos = cos[sp_rank * sl:(sp_rank + 1) * sl]
            sin = sin[sp_rank * sl:(sp_rank + 1) * sl]
        return cos, sin


class MultimodalRotaryEmbedding(torch.nn.Module):
    """3-section multimodal RoPE (reference rotary_pos_embedding.py:267
    MultimodalRotaryEmbedding, Qwen2-VL style): the d/2 frequency channels
    are split into ``mrope_section = [t, h, w]`` groups and each group's
    rotation angle is driven by the matching row of ``position_ids``
    [3, b, s] (temporal / height / width positions).  Tables stay
    host-precomputed; only the per-batch gather runs at step time."""

    def __init__(self, head_dim: int, mrope_section, theta: float = 10000.0):
        super().__init__()
        assert sum(mrope_section) == head_dim // 2, \
            f"mrope_section {mrope_section} must sum to head_dim/2"
        self.head_dim = head_dim
        self.theta = theta
        sect = torch.repeat_interleave(
            torch.arange(len(mrope_section)),
            torch.tensor(list(mrope_section)))
        self.register_buffer("section_of_channel", sect, persistent=False)

    def tables(self, position_ids: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
        """position_ids [3, b, s] int -> cos/sin [s, b, d/2] fp32."""
        assert position_ids.dim() == 3 and position_ids.shape[0] == 3
        d2 = self.head_dim // 2
        dev = position_ids.device
        inv_freq = 1.0 / (self.theta ** (
            torch.arange(0, self.head_dim, 2, device=dev).float()
            / self.head_dim))
        freqs = position_ids.float()[..., None] * inv_freq  # [3,b,s,d2]
        sel = self.section_of_channel.to(dev).view(1, 1, 1, d2) \
            .expand(1, *position_ids.shape[1:], d2)
        f = freqs.gather(0, sel).squeeze(0)                 # [b,s,d2]
        f = f.transpose(0, 1).contiguous()                  # [s,b,d2]
        return f.cos(), f.sin()


def apply_mrope_qk(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
                   sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k: [s, b, h, d]; cos/sin [s, b, d/2] (per-batch mrope tables).
    NEOX half-rotation, eager (mrope is not in the dense hot path)."""
    from ...ops.reference_ops import rope_apply_neox_batched
    return (rope_apply_neox_batched(q, cos, sin),
            rope_apply_neox_batched(k, cos, sin))


def zigzag_slice(x: torch.Tensor, cp_rank: int, cp_size: int) -> torch.Tensor:
    """Take chunks (cp_rank, 2cp-1-cp_rank) of 2cp equal chunks along dim 0
    (reference: redistribute.py:5-24 _zigzag_transformation)."""
    chunks = x.chunk(2 * cp_size, dim=0)
    return torch.cat([chunks[cp_rank], chunks[2 * cp_size - 1 - cp_rank]], dim=0)


def zigzag_unslice_index(cp_size: int) -> list:
    """Chunk order that reassembles a zigzag-sharded sequence: for gathered
    [rank0(lo,hi), rank1(lo,hi), ...] -> natural chunk order."""
    order = []
    pos = {}
    for r in range(cp_size):
        pos[r] = 2 * r          # rank r's low chunk position in gathered list
        pos[2 * cp_size - 1 - r] = 2 * r + 1
    for c in range(2 * cp_size):
        order.append(pos[c])
    return order


def apply_rope_qk(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
                  sin: torch.Tensor, interleaved: bool = False
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k: [s, b, h, d] SBH-heads layout. cos=None => no RoPE (learned
    positions, gpt family).  interleaved: GPT-J pairwise rotation
    (eager — the HIP kernel implements the NEOX-half layout; interleaved
    models are rare enough that the fused path stays NEOX-only)."""
    if cos is None:
        return q, k
    if interleaved:
        from ...ops.reference_ops import rope_apply_interleaved
        return (rope_apply_interleaved(q, cos, sin),
                rope_apply_interleaved(k, cos, sin))
    return apply_rope(q, cos, sin), apply_rope(k, cos, sin)
