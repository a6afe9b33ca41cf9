"""RoPE unit tests: seq-length interpolation + multimodal (mrope) sections.

Reference behaviors: rotary_pos_embedding.py rotary_seq_len_interpolation
factor (positions / factor) and MultimodalRotaryEmbedding:267 (3-section
channel split driven by position_ids [3,b,s]).
"""
import torch

from hetu_galvatron_amd.ops.reference_ops import rope_freqs, rope_apply
from hetu_galvatron_amd.runtime.transformer.rope import (
    MultimodalRotaryEmbedding, RotaryEmbedding, apply_mrope_qk)


def test_rope_interpolation_scales_positions():
    d = 64
    cos, sin = rope_freqs(32, d)
    cos4, sin4 = rope_freqs(128, d, interp=4.0)
    # position 4k under interp=4 rotates like position k unscaled
    torch.testing.assert_close(cos4[::4], cos, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(sin4[::4], sin, atol=1e-5, rtol=1e-5)


def test_rotary_embedding_wires_scaling():
    rot = RotaryEmbedding(64, scaling=2.0)
    base = RotaryEmbedding(64)
    cos_s, _ = rot.full_tables(16, torch.device("cpu"))
    cos_b, _ = base.full_tables(8, torch.device("cpu"))
    torch.testing.assert_close(cos_s[::2], cos_b, atol=1e-5, rtol=1e-5)


def test_mrope_equals_rope_for_text_positions():
    # all three position rows identical (pure text) => standard RoPE
    d, s, b, h = 64, 12, 2, 3
    rot = MultimodalRotaryEmbedding(d, [8, 12, 12])
    pos = torch.arange(s).view(1, 1, s).expand(3, b, s).contiguous()
    cos, sin = rot.tables(pos)                      # [s,b,d/2]
    cos_ref, sin_ref = rope_freqs(s, d)
    for bi in range(b):
        torch.testing.assert_close(cos[:, bi], cos_ref, atol=1e-5, rtol=1e-5)
    q = torch.randn(s, b, h, d)
    k = torch.randn(s, b, h, d)
    q2, k2 = apply_mrope_qk(q, k, cos, sin)
    torch.testing.assert_close(q2, rope_apply(q, cos_ref, sin_ref),
                               atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(k2, rope_apply(k, cos_ref, sin_ref),
                               atol=1e-5, rtol=1e-5)


def test_mrope_sections_use_their_own_positions():
    d = 32
    rot = MultimodalRotaryEmbedding(d, [4, 6, 6])
    s, b = 6, 1
    pos = torch.stack([torch.arange(s).view(1, s),
                       torch.zeros(1, s, dtype=torch.long),
                       torch.zeros(1, s, dtype=torch.long)])
    cos, _ = rot.tables(pos)                        # [s,1,16]
    # h/w channels (sections 1,2) see position 0 everywhere -> cos == 1
    torch.testing.assert_close(cos[:, 0, 4:], torch.ones(s, 12))
    # temporal channels rotate like standard rope's first 4 channels
    cos_ref, _ = rope_freqs(s, d)
    torch.testing.assert_close(cos[:, 0, :4], cos_ref[:, :4],
                               atol=1e-5, rtol=1e-5)


def test_mrope_section_must_cover_half_dim():
    import pytest
    with pytest.raises(AssertionError):
        MultimodalRotaryEmbedding(64, [8, 8, 8])


def test_rope_interleaved_semantics():
    """GPT-J pairwise rotation: pairs (2i, 2i+1) rotate by theta_i — the
    interleaved table equals NEOX applied to the de-interleaved tensor."""
    import torch
    from hetu_galvatron_amd.ops.reference_ops import (rope_apply,
                                                      rope_apply_interleaved,
                                                      rope_freqs)
    torch.manual_seed(0)
    s, b, h, d = 6, 2, 3, 16
    x = torch.randn(s, b, h, d)
    cos, sin = rope_freqs(s, d)
    got = rope_apply_interleaved(x, cos, sin)
    # de-interleave -> NEOX halves -> re-interleave
    perm = torch.cat([torch.arange(0, d, 2), torch.arange(1, d, 2)])
    inv = torch.argsort(perm)
    want = rope_apply(x[..., perm], cos, sin)[..., inv]
    torch.testing.assert_close(got, want, atol=1e-6, rtol=1e-6)
    # norm-preserving (pure rotation)
    torch.testing.assert_close(got.norm(), x.norm(), atol=1e-4, rtol=1e-5)


def test_rope_interleaved_end_to_end_flag():
    """rotary_interleaved must change the model's output (the flag is
    consumed, not dropped)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator

    losses = {}
    for flag in (False, True):
        cfg = load_config(base={
            "model": {"model_name": "tiny-llama",
                      "rotary_interleaved": flag},
            "train": {"global_train_batch_size": 2, "train_iters": 1,
                      "lr": 1e-3}})
        torch.manual_seed(0)
        m = GalvatronModel(cfg)
        torch.manual_seed(1)
        it = get_train_iterator(cfg, torch.device("cpu"))
        losses[flag] = m.forward_backward(next(it)).loss
    assert losses[False] != losses[True]
