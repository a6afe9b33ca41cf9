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
