"""Property-based tests (hypothesis) for the partitioning utilities that
every schedule keys off: microbatch chunking, pp layer division, zigzag
CP slicing, batch ramp-up.  Invariants hold for ALL inputs, not just the
shapes the example configs use."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from hetu_galvatron_amd.config.strategy import even_pp_division
from hetu_galvatron_amd.runtime.pipeline.engine import chunk_batch
from hetu_galvatron_amd.runtime.transformer.rope import (
    zigzag_slice, zigzag_unslice_index)


@settings(max_examples=200, deadline=None)
@given(rows=st.integers(1, 64), dp=st.integers(1, 8),
       chunks=st.integers(1, 16))
def test_chunk_batch_partitions_exactly(rows, dp, chunks):
    B = rows * dp
    ids = torch.arange(B).unsqueeze(-1).expand(B, 4)
    ctx = {"batch_size": B, "input_ids": ids, "labels": ids}
    mbs = chunk_batch(ctx, chunks, dp)
    # covers the batch exactly, in order, every piece a multiple of dp
    sizes = [m["input_ids"].shape[0] for m in mbs]
    assert sum(sizes) == B
    assert all(s % dp == 0 and s > 0 for s in sizes)
    assert len(mbs) == min(chunks, rows)
    # front-loaded remainder: sizes non-increasing, spread <= dp
    assert all(a >= b for a, b in zip(sizes, sizes[1:]))
    assert max(sizes) - min(sizes) <= dp
    cat = torch.cat([m["input_ids"] for m in mbs])
    assert torch.equal(cat, ids)


@settings(max_examples=200, deadline=None)
@given(n=st.integers(1, 256), pp=st.integers(1, 16))
def test_even_pp_division_properties(n, pp):
    div = even_pp_division(n, pp)
    assert sum(div) == n and len(div) == pp
    assert max(div) - min(div) <= 1
    # remainder goes to EARLY stages (warmup depth is deepest there)
    assert all(a >= b for a, b in zip(div, div[1:]))


@settings(max_examples=100, deadline=None)
@given(cp=st.integers(1, 8), unit=st.integers(1, 4))
def test_zigzag_slice_unslice_roundtrip(cp, unit):
    s = 2 * cp * unit
    x = torch.arange(s).unsqueeze(-1).float()
    shards = [zigzag_slice(x, r, cp) for r in range(cp)]
    # every rank holds chunks (r, 2cp-1-r): causal-balanced halves
    for r, sh in enumerate(shards):
        assert sh.shape[0] == 2 * unit
    gathered = torch.cat(shards)  # rank-major, as an allgather returns
    order = zigzag_unslice_index(cp)
    chunks = gathered.chunk(2 * cp)
    rebuilt = torch.cat([chunks[o] for o in order])
    assert torch.equal(rebuilt, x)


@settings(max_examples=100, deadline=None)
@given(unit=st.integers(1, 8), start_u=st.integers(1, 8),
       incr_u=st.integers(1, 8), steps=st.integers(0, 16),
       ramp=st.integers(1, 4096), samples=st.integers(0, 4096))
def test_batch_calculator_rampup_properties(unit, start_u, incr_u, steps,
                                            ramp, samples):
    # valid ramp spec by construction: target = start + steps*incr,
    # everything a multiple of micro_batch*dp (the calculator's contract)
    from hetu_galvatron_amd.runtime.optimizer.microbatches import (
        build_batch_calculator)
    start = start_u * unit
    incr = incr_u * unit
    target = start + steps * incr

    class T:
        global_train_batch_size = target
        rampup_batch_size = [start, incr, ramp]

    class C:
        train = T()

    calc = build_batch_calculator(C(), dp=unit, micro_batch_size=1)
    calc.update(samples)
    g = calc.get()[0]
    assert g % unit == 0           # always chunkable by the dp unit
    assert start <= g <= target    # monotone between start and target
    calc.update(10 ** 9)
    assert calc.get()[0] == target  # ramp completes


@settings(max_examples=25, deadline=None)
@given(st.lists(st.lists(st.integers(0, 2 ** 15 - 1), min_size=1,
                         max_size=50), min_size=1, max_size=20),
       st.sampled_from(["int32", "uint16", "int64"]))
def test_mmididx_roundtrip_random_docs(docs, dtype_name):
    """Megatron MMIDIDX writer -> reader identity for arbitrary corpora
    and every wire dtype (reference indexed_dataset.py format)."""
    import tempfile

    import numpy as np

    from hetu_galvatron_amd.runtime.datasets.indexed import (
        MegatronIndexedDataset, MegatronIndexedDatasetBuilder)
    dt = np.dtype(dtype_name)
    with tempfile.TemporaryDirectory() as td:
        prefix = f"{td}/corpus"
        b = MegatronIndexedDatasetBuilder(prefix, dtype=dt)
        for d in docs:
            b.add_document(np.asarray(d, dtype=dt))
        b.finalize()
        ds = MegatronIndexedDataset(prefix)
        assert len(ds) == len(docs)
        assert ds.doc_lens.tolist() == [len(d) for d in docs]
        for i, d in enumerate(docs):
            got = np.asarray(ds.doc(i))
            assert got.dtype == dt
            assert got.tolist() == d


@settings(max_examples=100, deadline=None)
@given(cp=st.sampled_from([1, 2, 4, 8]), tsp=st.sampled_from([1, 2, 4, 8]),
       unit=st.integers(1, 3))
def test_natural_rows_partition_sequence(cp, tsp, unit):
    """Every (cp_idx, tp_idx) owns a disjoint row set and the union is
    exactly [0, S) — the invariant redistribute() and the vocab-CE row
    slicing both build on."""
    from hetu_galvatron_amd.runtime.redistribute import natural_rows
    S = 2 * cp * tsp * unit
    seen = torch.zeros(S, dtype=torch.int32)
    for ci in range(cp):
        for ti in range(tsp):
            rows = natural_rows(S, cp, tsp, ci, ti, torch.device("cpu"))
            assert rows.shape[0] == S // (cp * tsp)
            seen[rows] += 1
    assert (seen == 1).all()
