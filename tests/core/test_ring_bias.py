"""Zigzag ring attention with additive bias (t5 relative bias under CP)
and bidirectional ring: world-2 ring == 1-process dense with the same
bias (reference role: attention_impl.py ring + relative bias, a gap in
the reference itself)."""
import pytest
import torch

TOL = 2e-4


def _dense_ref(q, k, v, bias, causal):
    """Full-sequence fp32 dense attention with bias; returns o."""
    from hetu_galvatron_amd.runtime.transformer.attention_impl import (
        eager_bias_attention)
    return eager_bias_attention(q, k, v, bias, causal, q.shape[-1] ** -0.5)


def _ring_worker(rank, world, causal):
    import torch.distributed as dist
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank, world_size=world)
    from hetu_galvatron_amd.runtime.transformer.attention_impl import (
        ZigzagRingAttention)
    from hetu_galvatron_amd.runtime.transformer.rope import zigzag_slice

    class G:  # minimal CommGroup-like wrapper over WORLD
        group = dist.group.WORLD
        ranks = list(range(world))
        size = world

        @staticmethod
        def index(r):
            return r

    torch.manual_seed(7)
    b, S, h, d = 2, 16, 2, 16
    q = torch.randn(b, S, h, d, dtype=torch.float32)
    k = torch.randn(b, S, h, d, dtype=torch.float32)
    v = torch.randn(b, S, h, d, dtype=torch.float32)
    bias = torch.randn(h, S, S) * 0.5
    bias_p = bias.clone().requires_grad_(True)
    do = torch.randn(b, S, h, d)

    # local zigzag shard (dim 1)
    def shard(t):
        return zigzag_slice(t.transpose(0, 1), rank, world).transpose(0, 1) \
            .contiguous()

    qs = shard(q).requires_grad_(True)
    ks = shard(k).requires_grad_(True)
    vs = shard(v).requires_grad_(True)
    rows = torch.cat([torch.arange(rank * S // 4, (rank + 1) * S // 4),
                      torch.arange((2 * world - 1 - rank) * S // 4,
                                   (2 * world - rank) * S // 4)])
    bias_local = bias_p[:, rows, :]
    ring = ZigzagRingAttention(G())
    o = ring(qs, ks, vs, causal=causal, softmax_scale=d ** -0.5,
             attn_bias=bias_local)
    o.backward(shard(do))

    # dense reference on the full sequence
    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    bf = bias.clone().requires_grad_(True)
    from hetu_galvatron_amd.runtime.transformer.attention_impl import (
        eager_bias_attention)
    of = eager_bias_attention(qf, kf, vf, bf, causal, d ** -0.5)
    of.backward(do)

    def err(a, b):
        return float((a - b).abs().max())

    # bias grad: this rank owns its q-row slice; sum over ranks == dense
    dbias_rows = bias_p.grad[:, rows, :]
    want_rows = bf.grad[:, rows, :]
    return {"o": err(o.detach(), shard(of.detach())),
            "dq": err(qs.grad, shard(qf.grad)),
            "dk": err(ks.grad, shard(kf.grad)),
            "dv": err(vs.grad, shard(vf.grad)),
            "dbias": err(dbias_rows, want_rows)}


@pytest.mark.distributed
@pytest.mark.parametrize("causal", [True, False])
def test_ring_bias_matches_dense(causal):
    from tests.utils import run_distributed
    res = run_distributed(_ring_worker, world_size=2, args=(causal,))
    for r in res:
        for k2, v2 in r.items():
            assert v2 < TOL, (k2, r)
