"""Indexed/blended dataset tests (C++ index builders)."""
import numpy as np
import pytest
import torch

from hetu_galvatron_amd.runtime.datasets import (
    BlendedDataset, GPTDataset, IndexedDataset, IndexedDatasetBuilder)


@pytest.fixture
def corpus(tmp_path):
    prefix = str(tmp_path / "corp")
    b = IndexedDatasetBuilder(prefix, dtype=np.int32)
    rng = np.random.RandomState(0)
    docs = [rng.randint(0, 1000, size=n) for n in (37, 120, 5, 260, 64)]
    for d in docs:
        b.add_document(d)
    b.finalize()
    return prefix, docs


def test_indexed_roundtrip(corpus):
    prefix, docs = corpus
    ds = IndexedDataset(prefix)
    assert len(ds) == len(docs)
    for i, d in enumerate(docs):
        assert np.array_equal(ds.doc(i), d)
    assert np.array_equal(ds.doc_lens, [len(d) for d in docs])


def test_megatron_mmididx_roundtrip(tmp_path, corpus):
    """Standard Megatron MMIDIDX corpora drop in: write a fixture in the
    exact MMapIndexedDataset byte layout (indexed_dataset.py:38-205),
    read it back through load_indexed_dataset, and feed GPTDataset —
    identical samples to the native-format corpus."""
    import struct
    from hetu_galvatron_amd.runtime.datasets import (
        MegatronIndexedDataset, MegatronIndexedDatasetBuilder,
        load_indexed_dataset)
    prefix, docs = corpus

    # hand-written fixture bytes (not via our builder) = the layout spec
    mp = str(tmp_path / "meg")
    sizes = [len(d) for d in docs]
    with open(mp + ".bin", "wb") as f:
        for d in docs:
            f.write(np.asarray(d, dtype=np.int32).tobytes())
    with open(mp + ".idx", "wb") as f:
        f.write(b"MMIDIDX\x00\x00")
        f.write(struct.pack("<Q", 1))
        f.write(struct.pack("<B", 4))  # int32
        f.write(struct.pack("<QQ", len(docs), len(docs)))
        f.write(np.asarray(sizes, dtype=np.int32).tobytes())
        ptrs, acc = [], 0
        for s in sizes:
            ptrs.append(acc)
            acc += s * 4
        f.write(np.asarray(ptrs, dtype=np.int64).tobytes())
        f.write(np.arange(len(docs), dtype=np.int64).tobytes())

    ds = load_indexed_dataset(mp)
    assert isinstance(ds, MegatronIndexedDataset)
    assert len(ds) == len(docs)
    for i, d in enumerate(docs):
        assert np.array_equal(ds.doc(i), d)
    assert np.array_equal(ds.doc_lens, sizes)

    # GPTDataset over the Megatron corpus == over the native corpus
    a = GPTDataset(load_indexed_dataset(prefix), seq_length=32,
                   num_samples=20, seed=7)
    b2 = GPTDataset(ds, seq_length=32, num_samples=20, seed=7)
    for i in range(20):
        assert np.array_equal(a[i], b2[i])

    # our builder's output re-reads through the same reader (export path)
    wp = str(tmp_path / "exp")
    wb = MegatronIndexedDatasetBuilder(wp, dtype=np.int32)
    for d in docs:
        wb.add_document(d)
    wb.finalize()
    ds2 = load_indexed_dataset(wp)
    for i, d in enumerate(docs):
        assert np.array_equal(ds2.doc(i), d)
    # byte-identical to the hand-written fixture
    assert open(wp + ".idx", "rb").read() == open(mp + ".idx", "rb").read()
    assert open(wp + ".bin", "rb").read() == open(mp + ".bin", "rb").read()


def test_gpt_dataset_samples(corpus):
    prefix, docs = corpus
    ds = GPTDataset(IndexedDataset(prefix), seq_length=32, num_samples=50,
                    seed=7)
    flat = np.concatenate(docs)
    seen = set()
    for i in range(50):
        s = ds[i]
        assert s.shape == (33,)
        seen.add(tuple(s[:4].tolist()))
        # each sample is a contiguous span of the wrapped corpus
        j = int(ds.shuffle_idx[i])
        doc, off = int(ds.sample_idx[j, 0]), int(ds.sample_idx[j, 1])
        start = sum(len(d) for d in docs[:doc]) + off
        want = np.concatenate([flat, flat])[start:start + 33]
        assert np.array_equal(s.numpy(), want)
    assert len(seen) > 10  # shuffled, varied
    # deterministic
    ds2 = GPTDataset(IndexedDataset(prefix), 32, 50, seed=7)
    assert torch.equal(ds[3], ds2[3])


def test_blended_proportions(corpus):
    prefix, _ = corpus
    d1 = GPTDataset(IndexedDataset(prefix), 16, 100, seed=1)
    d2 = GPTDataset(IndexedDataset(prefix), 16, 100, seed=2)
    bl = BlendedDataset([d1, d2], [0.75, 0.25], 400)
    counts = np.bincount(bl.dataset_index, minlength=2)
    assert abs(counts[0] - 300) <= 2 and abs(counts[1] - 100) <= 2
    _ = bl[0], bl[399]


def test_dataloader_megatron_path(tmp_path, corpus):
    prefix, _ = corpus
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import get_train_iterator
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 3},
        "data": {"dataset": "megatron", "data_path": [prefix]},
    })
    it = get_train_iterator(cfg, torch.device("cpu"))
    ctx = next(it)
    assert ctx["input_ids"].shape == (2, cfg.model.seq_length)


def test_t5_span_corruption_invariants():
    """Deterministic; sentinels pair up; content tokens preserved in order."""
    import numpy as np
    from hetu_galvatron_amd.runtime.datasets.t5_dataset import corrupt_spans
    V = 512
    toks = np.arange(100, 180)  # 80 distinct tokens, below sentinel range
    enc1, dec1 = corrupt_spans(toks, V, seed=5)
    enc2, dec2 = corrupt_spans(toks, V, seed=5)
    assert np.array_equal(enc1, enc2) and np.array_equal(dec1, dec2)
    sent_lo = V - 101
    enc_sent = [t for t in enc1 if t > sent_lo]
    dec_sent = [t for t in dec1 if t > sent_lo]
    # decoder has one extra final sentinel (EOS marker)
    assert len(dec_sent) == len(enc_sent) + 1
    assert enc_sent == dec_sent[:-1]
    # reconstruction: interleaving enc gaps with dec spans = original
    recon = []
    dec_pos = {}
    i = 0
    while i < len(dec1):
        s = dec1[i]; i += 1
        span = []
        while i < len(dec1) and dec1[i] <= sent_lo:
            span.append(dec1[i]); i += 1
        dec_pos[s] = span
    for t in enc1:
        if t > sent_lo:
            recon.extend(dec_pos[t])
        else:
            recon.append(t)
    assert np.array_equal(np.array(recon), toks)
    # noise density ~15%
    n_masked = sum(len(v) for v in dec_pos.values())
    assert 0.05 < n_masked / len(toks) < 0.35


def test_t5_masked_dataset_shapes():
    import torch
    from hetu_galvatron_amd.runtime.datasets.t5_dataset import T5MaskedDataset

    class Toy(torch.utils.data.Dataset):
        def __len__(self):
            return 4
        def __getitem__(self, i):
            return torch.randint(0, 300, (64,))

    ds = T5MaskedDataset(Toy(), enc_seq_len=72, dec_seq_len=32,
                         vocab_size=512)
    it = ds[0]
    assert it["enc_input_ids"].shape == (72,)
    assert it["dec_tokens"].shape == (33,)
    assert it["enc_input_ids"].max() < 512


def test_dataloader_t5_megatron_path(corpus):
    """t5 + megatron data: span-corruption pairs flow into the enc-dec
    batch context and train one step."""
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator
    prefix, _ = corpus
    cfg = load_config(base={
        "model": {"model_name": "tiny-t5"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"},
        "data": {"dataset": "megatron", "data_path": [prefix]},
    })
    it = get_train_iterator(cfg, torch.device("cpu"))
    ctx = next(it)
    assert ctx["enc_input_ids"].shape == (2, cfg.model.encoder_seq_length)
    assert ctx["input_ids"].shape == (2, cfg.model.seq_length)
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    st = model.forward_backward(ctx)
    assert st.loss > 0
