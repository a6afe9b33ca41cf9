"""HIP kernel numerics vs the plain-PyTorch fp32 reference (reference_ops).

All tests are @pytest.mark.gpu (MI355X box). Each kernel is compared
against the same-op fp32 torch reference on random data; asymmetric
operands everywhere so operand/output transposes cannot pass (guide
rule 16).
"""
import math

import pytest
import torch

from hetu_galvatron_amd.ops import reference_ops as ref

pytestmark = pytest.mark.gpu


def ext():
    from hetu_galvatron_amd.ops._ext import get_ext
    return get_ext(False)


def dev():
    return torch.device("cuda:0")


def assert_close(a, b, atol, rtol=2e-2, what=""):
    a = a.float().cpu()
    b = b.float().cpu()
    err = (a - b).abs().max().item()
    denom = b.abs().max().item() + 1e-6
    assert err <= atol + rtol * denom, \
        f"{what}: max|err|={err:.4e} vs atol={atol} rtol*max={rtol * denom:.4e}"


# ---------------------------------------------------------------------------
# MFMA layout probe: the fragment lane-map assumed by every MFMA kernel
# ---------------------------------------------------------------------------
def test_mfma_layout():
    torch.manual_seed(0)
    A = torch.randn(32, 16, device=dev()).bfloat16()
    B = torch.randn(16, 32, device=dev()).bfloat16()
    want = (A.float() @ B.float()).cpu()
    got_default = ext().mfma_probe(A, B, False).cpu()
    got_alt = ext().mfma_probe(A, B, True).cpu()
    err_d = (got_default - want).abs().max().item()
    err_a = (got_alt - want).abs().max().item()
    assert err_d < 0.1, (
        f"default MFMA A/B lane map wrong (err {err_d:.3e}); "
        f"alt map err {err_a:.3e} — if alt matches, switch mfma32_ab_k "
        f"to the two-block-of-4 form in ops/csrc/common.h")


# ---------------------------------------------------------------------------
# norms
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("shape", [(33, 4096), (4, 128, 1024), (7, 8192)])
def test_rmsnorm(dtype, shape):
    torch.manual_seed(1)
    x = torch.randn(shape, device=dev(), dtype=dtype)
    w = (torch.randn(shape[-1], device=dev(), dtype=dtype) * 0.1 + 1.0)
    y, inv = ext().rmsnorm_fwd(x, w, 1e-5)
    y_ref, inv_ref = ref.rmsnorm_fwd(x.float(), w.float(), 1e-5)
    atol = 3e-2 if dtype == torch.bfloat16 else 1e-5
    assert_close(y, y_ref, atol, what="rmsnorm y")
    assert_close(inv, inv_ref, 1e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="rmsnorm invrms")

    dy = torch.randn_like(x)
    dx, dw = ext().rmsnorm_bwd(dy.contiguous(), x, w, inv)
    dx_ref, dw_ref = ref.rmsnorm_bwd(dy.float(), x.float(), w.float(),
                                     inv_ref.float())
    assert_close(dx, dx_ref, 3e-2 if dtype == torch.bfloat16 else 1e-4,
                 what="rmsnorm dx")
    n_rows = x.numel() // shape[-1]
    assert_close(dw, dw_ref, (3e-2 if dtype == torch.bfloat16 else 1e-3) *
                 math.sqrt(n_rows), what="rmsnorm dw")


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_layernorm(dtype):
    torch.manual_seed(2)
    x = torch.randn(65, 2048, device=dev(), dtype=dtype)
    w = (torch.randn(2048, device=dev(), dtype=dtype) * 0.1 + 1.0)
    b = torch.randn(2048, device=dev(), dtype=dtype) * 0.1
    y, mean, inv = ext().layernorm_fwd(x, w, b, 1e-5)
    y_ref, mean_ref, inv_ref = ref.layernorm_fwd(x.float(), w.float(),
                                                 b.float(), 1e-5)
    atol = 3e-2 if dtype == torch.bfloat16 else 1e-5
    assert_close(y, y_ref, atol, what="ln y")
    assert_close(mean, mean_ref, 1e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="ln mean")

    dy = torch.randn_like(x)
    dx, dw, db = ext().layernorm_bwd(dy.contiguous(), x, w, mean, inv)
    dx_ref, dw_ref, db_ref = ref.layernorm_bwd(dy.float(), x.float(),
                                               w.float(), mean_ref.float(),
                                               inv_ref.float())
    assert_close(dx, dx_ref, 3e-2 if dtype == torch.bfloat16 else 1e-4,
                 what="ln dx")
    assert_close(dw, dw_ref, 0.3 if dtype == torch.bfloat16 else 1e-2,
                 what="ln dw")
    assert_close(db, db_ref, 0.3 if dtype == torch.bfloat16 else 1e-2,
                 what="ln db")


# ---------------------------------------------------------------------------
# swiglu / rope
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_swiglu(dtype):
    torch.manual_seed(3)
    x = torch.randn(17, 512, device=dev(), dtype=dtype)
    y = ext().swiglu_fwd(x)
    y_ref = ref.swiglu_fwd(x.float())
    assert_close(y, y_ref, 2e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="swiglu y")
    dy = torch.randn_like(y)
    dx = ext().swiglu_bwd(dy.contiguous(), x)
    dx_ref = ref.swiglu_bwd(dy.float(), x.float())
    assert_close(dx, dx_ref, 2e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="swiglu dx")


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_rope(dtype):
    torch.manual_seed(4)
    s, b, h, d = 128, 2, 4, 128
    x = torch.randn(s, b, h, d, device=dev(), dtype=dtype)
    cos, sin = ref.rope_freqs(s, d, device=dev())
    y = ext().rope_fwd(x, cos, sin, False)
    y_ref = ref.rope_apply(x.float(), cos, sin)
    assert_close(y, y_ref, 2e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="rope y")
    # conj is the inverse rotation
    back = ext().rope_fwd(y, cos, sin, True)
    assert_close(back, x.float(),
                 5e-2 if dtype == torch.bfloat16 else 1e-5, what="rope inv")


# ---------------------------------------------------------------------------
# flash attention
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("d", [64, 128])
@pytest.mark.parametrize("causal", [True, False])
@pytest.mark.parametrize("hq,hkv", [(8, 8), (8, 2)])
@pytest.mark.parametrize("sq", [256, 300])
def test_flash_fwd(d, causal, hq, hkv, sq):
    torch.manual_seed(5)
    b = 2
    q = torch.randn(b, sq, hq, d, device=dev()).bfloat16()
    k = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    v = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, causal, scale)
    o_ref, lse_ref = ref.attention_fwd(q.float(), k.float(), v.float(),
                                       causal, scale)
    assert_close(o, o_ref, 3e-2, what=f"flash o d{d} c{causal}")
    assert_close(lse, lse_ref, 2e-2, what="flash lse")


@pytest.mark.parametrize("d", [64, 128])
@pytest.mark.parametrize("hq,hkv", [(4, 4), (4, 1)])
def test_flash_bwd(d, hq, hkv):
    torch.manual_seed(6)
    b, sq = 2, 192
    q = torch.randn(b, sq, hq, d, device=dev()).bfloat16()
    k = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    v = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale)
    do = torch.randn_like(o)
    dq, dk, dv = ext().flash_attn_bwd(do.contiguous(), q, k, v, o, lse,
                                      True, scale)
    dq_ref, dk_ref, dv_ref = ref.attention_bwd(do.float(), q.float(),
                                               k.float(), v.float(), None,
                                               None, True, scale)
    assert_close(dq, dq_ref, 6e-2, rtol=3e-2, what="flash dq")
    assert_close(dk, dk_ref, 6e-2, rtol=3e-2, what="flash dk")
    assert_close(dv, dv_ref, 6e-2, rtol=3e-2, what="flash dv")


def test_flash_sbhd_layout():
    """sbhd=True ([s,b,h,d], the runtime's native layout) must match the
    bshd path bit-for-bit on the same data, fwd + bwd."""
    torch.manual_seed(17)
    b, sq, hq, hkv, d = 3, 512, 8, 2, 128
    q = torch.randn(sq, b, hq, d, device=dev()).bfloat16()
    k = torch.randn(sq, b, hkv, d, device=dev()).bfloat16()
    v = torch.randn(sq, b, hkv, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale, None, True)
    qb = q.permute(1, 0, 2, 3).contiguous()
    kb = k.permute(1, 0, 2, 3).contiguous()
    vb = v.permute(1, 0, 2, 3).contiguous()
    ob, lseb = ext().flash_attn_fwd(qb, kb, vb, True, scale)
    assert torch.equal(o.permute(1, 0, 2, 3), ob), "sbhd fwd o mismatch"
    assert torch.equal(lse, lseb), "sbhd lse mismatch"
    do = torch.randn_like(o)
    dq, dk, dv = ext().flash_attn_bwd(do, q, k, v, o, lse, True, scale,
                                      None, True)
    dqb, dkb, dvb = ext().flash_attn_bwd(
        do.permute(1, 0, 2, 3).contiguous(), qb, kb, vb, ob, lseb, True,
        scale)
    assert torch.equal(dq.permute(1, 0, 2, 3), dqb), "sbhd dq"
    assert torch.equal(dk.permute(1, 0, 2, 3), dkb), "sbhd dk"
    assert torch.equal(dv.permute(1, 0, 2, 3), dvb), "sbhd dv"


def test_flash_bench_shape_numerics():
    """Bench-shape insurance: seq-4096 GQA 32/8 fwd+bwd vs
    the fp32 reference — kernel regressions can't hide behind toy shapes.
    Uses torch sdpa as the oracle for the forward (fp32) and autograd for
    the backward; one (b=1) slice of the bench microbatch, <30 s."""
    torch.manual_seed(11)
    b, sq, hq, hkv, d = 1, 4096, 32, 8, 128
    q = torch.randn(b, sq, hq, d, device=dev()).bfloat16()
    k = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    v = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale)
    # fp32 sdpa oracle in [b, h, s, d] with GQA expansion
    q32 = q.float().permute(0, 2, 1, 3)
    k32 = k.float().permute(0, 2, 1, 3).repeat_interleave(hq // hkv, dim=1)
    v32 = v.float().permute(0, 2, 1, 3).repeat_interleave(hq // hkv, dim=1)
    q32.requires_grad_(True); k32.requires_grad_(True); v32.requires_grad_(True)
    o_ref = torch.nn.functional.scaled_dot_product_attention(
        q32, k32, v32, is_causal=True, scale=scale)
    assert_close(o, o_ref.permute(0, 2, 1, 3), 3e-2, what="bench-shape o")
    do = torch.randn_like(o)
    o_ref.backward(do.float().permute(0, 2, 1, 3))
    dq, dk, dv = ext().flash_attn_bwd(do.contiguous(), q, k, v, o, lse,
                                      True, scale)
    g = hq // hkv
    dk_ref = k32.grad.view(b, hkv, g, sq, d).sum(2).permute(0, 2, 1, 3)
    dv_ref = v32.grad.view(b, hkv, g, sq, d).sum(2).permute(0, 2, 1, 3)
    assert_close(dq, q32.grad.permute(0, 2, 1, 3), 6e-2, rtol=3e-2,
                 what="bench-shape dq")
    assert_close(dk, dk_ref, 1.2e-1, rtol=3e-2, what="bench-shape dk")
    assert_close(dv, dv_ref, 6e-2, rtol=3e-2, what="bench-shape dv")


@pytest.mark.parametrize("causal", [True, False])
@pytest.mark.parametrize("sq,skv", [(256, 256), (128, 256)])
def test_flash_bias_fwd_bwd(causal, sq, skv):
    """Biased flash (t5 relative bias): o/lse/dq/dk/dv/dbias vs the fp32
    reference (the round-2 native-bias kernel item)."""
    torch.manual_seed(12)
    b, h, d = 2, 4, 64
    q = torch.randn(b, sq, h, d, device=dev()).bfloat16()
    k = torch.randn(b, skv, h, d, device=dev()).bfloat16()
    v = torch.randn(b, skv, h, d, device=dev()).bfloat16()
    bias = (torch.randn(h, sq, skv, device=dev()) * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, causal, scale, bias)
    o_ref, lse_ref = ref.attention_fwd(q.float(), k.float(), v.float(),
                                       causal, scale, bias.float())
    assert_close(o, o_ref, 3e-2, what=f"bias o c{causal}")
    assert_close(lse, lse_ref, 2e-2, what="bias lse")
    do = torch.randn_like(o)
    dq, dk, dv, dbias = ext().flash_attn_bwd(do.contiguous(), q, k, v, o,
                                             lse, causal, scale, bias)
    dq_r, dk_r, dv_r, db_r = ref.attention_bwd(
        do.float(), q.float(), k.float(), v.float(), None, None, causal,
        scale, bias.float())
    assert_close(dq, dq_r, 6e-2, rtol=3e-2, what="bias dq")
    assert_close(dk, dk_r, 6e-2, rtol=3e-2, what="bias dk")
    assert_close(dv, dv_r, 6e-2, rtol=3e-2, what="bias dv")
    assert_close(dbias, db_r, 6e-2, rtol=3e-2, what="dbias")


def test_flash_bias_gqa_d128():
    torch.manual_seed(13)
    b, sq, hq, hkv, d = 1, 192, 4, 2, 128
    q = torch.randn(b, sq, hq, d, device=dev()).bfloat16()
    k = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    v = torch.randn(b, sq, hkv, d, device=dev()).bfloat16()
    bias = (torch.randn(hq, sq, sq, device=dev()) * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale, bias)
    o_ref, _ = ref.attention_fwd(q.float(), k.float(), v.float(), True,
                                 scale, bias.float())
    assert_close(o, o_ref, 3e-2, what="bias gqa o")
    do = torch.randn_like(o)
    dq, dk, dv, dbias = ext().flash_attn_bwd(do.contiguous(), q, k, v, o,
                                             lse, True, scale, bias)
    dq_r, dk_r, dv_r, db_r = ref.attention_bwd(
        do.float(), q.float(), k.float(), v.float(), None, None, True,
        scale, bias.float())
    assert_close(dq, dq_r, 6e-2, rtol=3e-2, what="bias gqa dq")
    assert_close(dk, dk_r, 6e-2, rtol=3e-2, what="bias gqa dk")
    assert_close(dbias, db_r, 6e-2, rtol=3e-2, what="bias gqa dbias")


def test_flash_cross_lengths():
    """sq != skv (ring-CP block form), bottom-right causal alignment."""
    torch.manual_seed(7)
    b, hq, d = 1, 4, 128
    q = torch.randn(b, 128, hq, d, device=dev()).bfloat16()
    k = torch.randn(b, 256, hq, d, device=dev()).bfloat16()
    v = torch.randn(b, 256, hq, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale)
    o_ref, lse_ref = ref.attention_fwd(q.float(), k.float(), v.float(),
                                       True, scale)
    assert_close(o, o_ref, 3e-2, what="flash cross o")
    assert_close(lse, lse_ref, 2e-2, what="flash cross lse")


# ---------------------------------------------------------------------------
# cross entropy
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("V", [1000, 16032])
def test_vocab_ce(dtype, V):
    torch.manual_seed(8)
    n, vocab_start = 64, 2 * V
    logits = torch.randn(n, V, device=dev(), dtype=dtype) * 4
    # half the targets on-shard, half off
    target = torch.randint(0, 4 * V, (n,), device=dev())
    gmax = ext().ce_max(logits)
    gmax_ref = logits.float().max(dim=-1).values
    assert_close(gmax, gmax_ref, 1e-3, what="ce max")
    sumexp, tlogit = ext().ce_sum_target(logits, target, gmax_ref.contiguous(),
                                         vocab_start)
    se_ref, tl_ref = ref.vocab_ce_fwd_local(logits.float(), target, gmax_ref,
                                            vocab_start, vocab_start + V)
    assert_close(sumexp, se_ref, 1e-2, what="ce sumexp")
    assert_close(tlogit, tl_ref, 1e-3, what="ce tlogit")

    gout = torch.randn(n, device=dev()).float()
    dl_ref = ref.vocab_ce_bwd_local(logits.float(), target, gmax_ref, se_ref,
                                    gout, vocab_start, vocab_start + V)
    dl = ext().ce_bwd(logits.clone(), target, gmax_ref.contiguous(),
                      se_ref.contiguous(), gout, vocab_start)
    assert_close(dl, dl_ref, 2e-2 if dtype == torch.bfloat16 else 1e-5,
                 what="ce dlogits")


# ---------------------------------------------------------------------------
# fused adamw
# ---------------------------------------------------------------------------
def test_fused_adamw():
    torch.manual_seed(9)
    sizes = [1000, 4097, 31]
    masters = [torch.randn(s, device=dev()).float() for s in sizes]
    grads = [torch.randn(s, device=dev()).bfloat16() for s in sizes]
    ms = [torch.rand(s, device=dev()).float() * 0.1 for s in sizes]
    vs = [torch.rand(s, device=dev()).float() * 0.01 for s in sizes]
    outs = [torch.zeros(s, device=dev()).bfloat16() for s in sizes]

    masters_ref = [t.clone() for t in masters]
    ms_ref = [t.clone() for t in ms]
    vs_ref = [t.clone() for t in vs]
    outs_ref = [t.clone() for t in outs]

    ext().fused_adamw(masters, grads, ms, vs, outs, 3, 1e-3, 0.9, 0.95,
                      1e-8, 0.01)
    ref.adamw_step(outs_ref, grads, ms_ref, vs_ref, masters_ref, 3, 1e-3,
                   0.9, 0.95, 1e-8, 0.01)
    for i in range(len(sizes)):
        assert_close(masters[i], masters_ref[i], 1e-5, what=f"adam master {i}")
        assert_close(ms[i], ms_ref[i], 1e-5, what=f"adam m {i}")
        assert_close(vs[i], vs_ref[i], 1e-6, what=f"adam v {i}")
        assert_close(outs[i], outs_ref[i], 1e-2, what=f"adam out {i}")


def test_add_rmsnorm_fused():
    """Fused residual-add + rmsnorm vs the unfused composition, fwd+bwd
    (incl. the dsum fold in backward)."""
    torch.manual_seed(16)
    n, H = 1024, 4096
    x = torch.randn(n, H, device=dev()).bfloat16().requires_grad_(True)
    r = torch.randn(n, H, device=dev()).bfloat16().requires_grad_(True)
    w = torch.randn(H, device=dev()).bfloat16().requires_grad_(True)
    from hetu_galvatron_amd.ops import fused_add_rms_norm, rms_norm
    y, s = fused_add_rms_norm(x, r, w, 1e-5)
    x2 = x.detach().requires_grad_(True)
    r2 = r.detach().requires_grad_(True)
    w2 = w.detach().requires_grad_(True)
    s2 = x2 + r2
    y2 = rms_norm(s2, w2, 1e-5)
    assert_close(y, y2, 3e-2, what="addnorm y")
    assert_close(s, s2, 1e-6, what="addnorm sum")
    dy = torch.randn_like(y)
    ds = torch.randn_like(s)
    (y.float() * dy.float()).sum().backward(retain_graph=True)
    (s.float() * ds.float()).sum().backward()
    (y2.float() * dy.float()).sum().backward(retain_graph=True)
    (s2.float() * ds.float()).sum().backward()
    assert_close(x.grad, x2.grad, 3e-2, what="addnorm dx")
    assert_close(r.grad, r2.grad, 3e-2, what="addnorm dres")
    assert_close(w.grad, w2.grad, 1e-1, rtol=3e-2, what="addnorm dw")


def test_fused_adamw_gscale():
    """gscale folds the clip factor into the kernel: equals pre-scaling
    the grads then running with gscale=1."""
    torch.manual_seed(14)
    n = 5000
    master = torch.randn(n, device=dev()).float()
    g = torch.randn(n, device=dev()).bfloat16()
    m = torch.rand(n, device=dev()).float() * 0.1
    v = torch.rand(n, device=dev()).float() * 0.01
    out = torch.zeros(n, device=dev()).bfloat16()
    m2, v2 = m.clone(), v.clone()
    master2, out2 = master.clone(), out.clone()
    s = 0.37
    ext().fused_adamw([master], [g], [m], [v], [out], 2, 1e-3, 0.9, 0.95,
                      1e-8, 0.01, s)
    ext().fused_adamw([master2], [(g.float() * s).bfloat16()], [m2], [v2],
                      [out2], 2, 1e-3, 0.9, 0.95, 1e-8, 0.01, 1.0)
    # bf16 re-quantization of the pre-scaled grads is the only difference
    assert_close(master, master2, 2e-3, what="gscale master")
    assert_close(m, m2, 2e-3, what="gscale m")


def test_multi_sumsq():
    torch.manual_seed(15)
    xs = [torch.randn(n, device=dev()).float()
          for n in (17, 4096, 1 << 21, 123457)]
    got = ext().multi_sumsq(xs)[0]
    want = sum((x.double() ** 2).sum() for x in xs)
    assert abs(float(got) - float(want)) / float(want) < 1e-5, (got, want)


def test_grad_accum():
    torch.manual_seed(10)
    flat = torch.randn(10000, device=dev()).float()
    base = flat.clone()
    g = torch.randn(4097, device=dev()).bfloat16()
    ext().grad_accum(flat, g, 123)
    want = base.clone()
    want[123:123 + 4097] += g.float()
    assert_close(flat, want, 1e-5, what="grad_accum")


@pytest.mark.parametrize("counts", [[100, 28, 0, 130], [128, 128, 128, 128]])
def test_grouped_gemm(counts):
    from hetu_galvatron_amd.ops.functional import grouped_gemm
    torch.manual_seed(11)
    E, K, N = 4, 256, 384
    M = sum(counts)
    a = (torch.randn(M, K, device=dev()) / 8).bfloat16().requires_grad_(True)
    w = (torch.randn(E, K, N, device=dev()) / 8).bfloat16().requires_grad_(True)
    c = grouped_gemm(a, w, counts)
    # reference: per-expert mm in fp32
    a32 = a.detach().float().requires_grad_(True)
    w32 = w.detach().float().requires_grad_(True)
    outs, s = [], 0
    for e, m in enumerate(counts):
        outs.append(a32[s:s + m] @ w32[e])
        s += m
    c_ref = torch.cat(outs)
    assert_close(c, c_ref, 5e-2, what="grouped_gemm fwd")
    dc = torch.randn_like(c)
    c.backward(dc)
    c_ref.backward(dc.float())
    assert_close(a.grad, a32.grad, 6e-2, rtol=3e-2, what="grouped_gemm dA")
    assert_close(w.grad, w32.grad, 6e-2, rtol=3e-2, what="grouped_gemm dW")


def test_moe_permute_unpermute():
    from hetu_galvatron_amd.ops.functional import moe_permute, moe_unpermute
    torch.manual_seed(12)
    n, h, E, k = 64, 128, 4, 2
    x = torch.randn(n, h, device=dev()).bfloat16().requires_grad_(True)
    idx = torch.randint(0, E, (n, k), device=dev())
    probs = torch.softmax(torch.randn(n, k, device=dev()), -1).reshape(-1)
    flat = idx.reshape(-1)
    order = torch.argsort(flat, stable=True)
    rows = order // k
    perm = moe_permute(x, rows)
    assert torch.equal(perm.detach(), x.detach()[rows])
    p_sorted = probs[order]
    out = moe_unpermute(perm, p_sorted, order, n, k)
    # reference (pure torch)
    x2 = x.detach().clone().requires_grad_(True)
    perm2 = x2[rows]
    full = perm2.new_zeros(n * k, h)
    full[order] = perm2 * p_sorted.unsqueeze(-1).to(perm2.dtype)
    out_ref = full.reshape(n, k, h).sum(1)
    assert_close(out, out_ref, 2e-2, what="moe unpermute fwd")
    g = torch.randn_like(out)
    out.backward(g)
    out_ref.backward(g)
    assert_close(x.grad, x2.grad, 3e-2, what="moe permute/unpermute grad")


# ---------------------------------------------------------------------------
# decode attention (serving)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("d,hq,hkv,cur", [(128, 32, 8, 1), (128, 32, 8, 777),
                                          (64, 8, 8, 4096), (128, 16, 2, 300)])
def test_decode_attn(d, hq, hkv, cur):
    from hetu_galvatron_amd.ops import decode_attention
    from hetu_galvatron_amd.ops._ext import get_ext
    torch.manual_seed(0)
    b, max_s = 2, max(cur, 512)
    q = torch.randn(b, hq, d, device="cuda", dtype=torch.bfloat16)
    kc = torch.randn(b, max_s, hkv, d, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(b, max_s, hkv, d, device="cuda", dtype=torch.bfloat16)
    o = get_ext().decode_attn(q, kc, vc, cur, 1.0 / math.sqrt(d))
    # fp32 dense reference
    k = kc[:, :cur].float().repeat_interleave(hq // hkv, dim=2)
    v = vc[:, :cur].float().repeat_interleave(hq // hkv, dim=2)
    att = torch.einsum("bhd,bshd->bhs", q.float(), k) / math.sqrt(d)
    want = torch.einsum("bhs,bshd->bhd", att.softmax(-1), v)
    assert (o.float() - want).abs().max() < 0.02


def test_generate_gpu_native_decode():
    """End-to-end generation on GPU exercises the native decode kernel and
    matches full-prefix recompute token-for-token."""
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.inference import (GalvatronGenerator,
                                                      KVCache)
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 1},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    ids = torch.randint(0, cfg.model.vocab_size, (2, 9), device="cuda")
    out = gen.generate(ids, max_new_tokens=5, temperature=0.0)
    assert out.shape == (2, 14)
    # each generated token must be a near-argmax of a full-prefix recompute
    # (bf16: the two attention kernels may differ in last-bit rounding)
    for i in range(5):
        cache = KVCache(len(gen.layers), 2, 64, cfg.model.kv_heads,
                        cfg.model.head_dim, ids.device)
        logits = gen._forward_tokens(out[:, :9 + i], cache)
        got = logits.gather(1, out[:, 9 + i:10 + i])
        assert (logits.max(-1, keepdim=True).values - got).max() < 0.05


def test_decode_attn_graph_variant():
    """Device-resident length (hipGraph mode) == host-length kernel."""
    from hetu_galvatron_amd.ops._ext import get_ext
    torch.manual_seed(3)
    b, hq, hkv, d, max_s = 2, 32, 8, 128, 2048
    q = torch.randn(b, hq, d, device="cuda", dtype=torch.bfloat16)
    kc = torch.randn(b, max_s, hkv, d, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(b, max_s, hkv, d, device="cuda", dtype=torch.bfloat16)
    for cur in (1, 300, 2048):
        cur32 = torch.tensor([cur], dtype=torch.int32, device="cuda")
        og = get_ext().decode_attn_graph(q, kc, vc, cur32, max_s,
                                         d ** -0.5)
        k = kc[:, :cur].float().repeat_interleave(hq // hkv, dim=2)
        v = vc[:, :cur].float().repeat_interleave(hq // hkv, dim=2)
        att = torch.einsum("bhd,bshd->bhs", q.float(), k) * d ** -0.5
        want = torch.einsum("bhs,bshd->bhd", att.softmax(-1), v)
        assert (og.float() - want).abs().max() < 0.02, cur


def test_generate_graphed_gpu():
    """hipGraph-captured decode: every token a near-argmax of a
    full-prefix recompute (falls back to eager if capture unavailable)."""
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.inference import (GalvatronGenerator,
                                                      KVCache)
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 1},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    ids = torch.randint(0, cfg.model.vocab_size, (2, 9), device="cuda")
    out = gen.generate_graphed(ids, max_new_tokens=8, warmup_steps=2)
    assert out.shape == (2, 17)
    for i in range(8):
        cache = KVCache(len(gen.layers), 2, 64, cfg.model.kv_heads,
                        cfg.model.head_dim, ids.device)
        logits = gen._forward_tokens(out[:, :9 + i], cache)
        got = logits.gather(1, out[:, 9 + i:10 + i])
        assert (logits.max(-1, keepdim=True).values - got).max() < 0.05, i


def test_chunked_prefill_gpu():
    """Chunked prefill == one-shot prefill on the native kernels (bf16)."""
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.inference import (GalvatronGenerator,
                                                      KVCache)
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 1},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    ids = torch.randint(0, cfg.model.vocab_size, (2, 12), device="cuda")
    mk = lambda: KVCache(len(gen.layers), 2, 64, cfg.model.kv_heads,
                         cfg.model.head_dim, ids.device)
    c1, c2 = mk(), mk()
    full = gen._forward_tokens(ids, c1)
    gen._forward_tokens(ids[:, :5], c2)
    chunked = gen._forward_tokens(ids[:, 5:], c2)
    assert (full - chunked).abs().max() < 0.1  # bf16 path tolerance


@pytest.mark.parametrize("window", [64, 200])
@pytest.mark.parametrize("sq", [256, 512])
def test_flash_sliding_window(window, sq):
    """Native mistral-style sliding window vs the fp32 reference
    (keys visible to q: [q-window+1, q])."""
    torch.manual_seed(18)
    b, h, d = 2, 4, 128
    q = torch.randn(b, sq, h, d, device=dev()).bfloat16()
    k = torch.randn(b, sq, h, d, device=dev()).bfloat16()
    v = torch.randn(b, sq, h, d, device=dev()).bfloat16()
    scale = 1.0 / math.sqrt(d)
    o, lse = ext().flash_attn_fwd(q, k, v, True, scale, None, False, window)
    o_ref, lse_ref = ref.attention_fwd(q.float(), k.float(), v.float(),
                                       True, scale, window=window)
    assert_close(o, o_ref, 3e-2, what=f"win{window} o")
    assert_close(lse, lse_ref, 2e-2, what="win lse")
    do = torch.randn_like(o)
    dq, dk, dv = ext().flash_attn_bwd(do.contiguous(), q, k, v, o, lse,
                                      True, scale, None, False, window)
    dq_r, dk_r, dv_r = ref.attention_bwd(do.float(), q.float(), k.float(),
                                         v.float(), None, None, True,
                                         scale, window=window)
    assert_close(dq, dq_r, 6e-2, rtol=3e-2, what="win dq")
    assert_close(dk, dk_r, 6e-2, rtol=3e-2, what="win dk")
    assert_close(dv, dv_r, 6e-2, rtol=3e-2, what="win dv")


def test_decode_attn_windowed():
    """decode_attn(start=): pointer-offset windowing == dense reference
    over the window slice (incl. the split-KV path at small b*hq)."""
    torch.manual_seed(19)
    b, hq, hkv, d, S, W = 1, 4, 2, 128, 4096, 1024
    q = torch.randn(b, hq, d, device=dev()).bfloat16()
    kc = torch.randn(b, S, hkv, d, device=dev()).bfloat16()
    vc = torch.randn(b, S, hkv, d, device=dev()).bfloat16()
    scale = d ** -0.5
    o = ext().decode_attn(q, kc, vc, S, scale, S - W)
    k = kc[:, S - W:S].float().repeat_interleave(hq // hkv, dim=2)
    v = vc[:, S - W:S].float().repeat_interleave(hq // hkv, dim=2)
    att = torch.einsum("bhd,bshd->bhs", q.float(), k) * scale
    want = torch.einsum("bhs,bshd->bhd", att.softmax(-1), v)
    assert_close(o, want, 3e-2, what="windowed decode")
