"""T5 bucketized relative-position bias units (family feature the dist
t5 tests exercise end-to-end; reference parity: HF T5 relative_attention_bias
behavior re-derived from the T5 paper bucketing)."""
import torch

from hetu_galvatron_amd.runtime.transformer.attention import (
    eager_bias_attention)
from hetu_galvatron_amd.runtime.transformer.relative_bias import (
    RelativePositionBias, t5_relative_bucket)


def test_bucket_function_properties():
    q = torch.arange(64)
    rel = q[None, :] - q[:, None]  # memory - query
    bi = t5_relative_bucket(rel, True, 32, 128)
    ca = t5_relative_bucket(rel, False, 32, 128)
    assert int(bi.max()) < 32 and int(ca.max()) < 32
    assert int(bi.min()) >= 0 and int(ca.min()) >= 0
    # causal: all future (rel > 0) collapse to bucket 0
    assert (ca[rel > 0] == 0).all()
    # bidirectional: past and future use disjoint bucket halves
    assert (bi[rel > 0] >= 16).all() and (bi[rel < 0] < 16).all()
    # exact small offsets get unique buckets
    assert ca[5, 4] != ca[5, 3] != ca[5, 2]
    # monotone decay in distance: bucket ids non-decreasing with |offset|
    row = ca[63]
    assert (row[:-1].flip(0).diff() >= 0).all()


def test_bias_head_slicing_matches_full():
    torch.manual_seed(0)
    rb = RelativePositionBias(32, 128, 8, bidirectional=True)
    full = rb(16, 16, torch.device("cpu"))
    assert full.shape == (8, 16, 16)
    halves = torch.cat([rb(16, 16, torch.device("cpu"), 0, 4),
                        rb(16, 16, torch.device("cpu"), 4, 8)])
    assert torch.equal(full, halves)


def test_eager_bias_attention_matches_dense():
    torch.manual_seed(1)
    b, sq, hq, hkv, d = 2, 8, 4, 2, 16
    q = torch.randn(b, sq, hq, d)
    k = torch.randn(b, sq, hkv, d)
    v = torch.randn(b, sq, hkv, d)
    bias = torch.randn(hq, sq, sq)
    out = eager_bias_attention(q, k, v, bias, causal=True, scale=0.25)
    ke = k.repeat_interleave(2, dim=2)
    ve = v.repeat_interleave(2, dim=2)
    for bi in range(b):
        for h in range(hq):
            s = q[bi, :, h] @ ke[bi, :, h].T * 0.25 + bias[h]
            s = s.masked_fill(torch.ones(sq, sq, dtype=torch.bool).triu(1),
                              float("-inf"))
            o = s.softmax(-1) @ ve[bi, :, h]
            assert torch.allclose(out[bi, :, h], o, atol=1e-5)


def test_bias_zero_matches_flash_reference():
    """bias==0 must reduce to the plain attention path."""
    from hetu_galvatron_amd.ops.reference_ops import attention_fwd
    torch.manual_seed(2)
    b, sq, h, d = 2, 8, 2, 16
    q = torch.randn(b, sq, h, d)
    k = torch.randn(b, sq, h, d)
    v = torch.randn(b, sq, h, d)
    out = eager_bias_attention(q, k, v, torch.zeros(h, sq, sq), True,
                               d ** -0.5)
    want, _ = attention_fwd(q, k, v, causal=True)
    assert torch.allclose(out, want, atol=1e-5)
