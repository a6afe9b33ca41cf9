"""GPT pretraining dataset over indexed token stores (+ blending).

Reference: galvatron/core/runtime/datasets/megatron/gpt_dataset.py and
blended_megatron_dataset_*.py — sample index built by the C++ helper
(csrc_cpu/dataset_helpers.cpp; reference helpers.cpp:143 build_sample_idx,
:75 build_blending_indices), deterministic shuffle, weighted blending.
"""
from __future__ import annotations

from typing import List, Sequence

import numpy as np
import torch

from .indexed import IndexedDataset, load_indexed_dataset


def _helpers():
    import torch  # noqa: F401 — extension links torch libs
    from ... import _galvatron_dataset_helpers as h
    return h


class GPTDataset(torch.utils.data.Dataset):
    """Fixed-length causal-LM samples (seq_length+1 tokens) over a document
    store; deterministic (seeded) shuffle; epochs wrap."""

    def __init__(self, indexed: IndexedDataset, seq_length: int,
                 num_samples: int, seed: int = 1234, shuffle: bool = True):
        self.indexed = indexed
        self.seq_length = seq_length
        self.num_samples = num_samples
        h = _helpers()
        self.sample_idx = h.build_sample_idx(indexed.doc_lens,
                                             seq_length, num_samples)
        self.shuffle_idx = (h.build_shuffle_idx(num_samples, seed)
                            if shuffle else np.arange(num_samples))

    def __len__(self) -> int:
        return self.num_samples

    def __getitem__(self, i: int) -> torch.Tensor:
        s = int(self.shuffle_idx[i % self.num_samples])
        doc, off = int(self.sample_idx[s, 0]), int(self.sample_idx[s, 1])
        span = self.indexed.read_span(doc, off, self.seq_length + 1)
        return torch.from_numpy(span)


class BlendedDataset(torch.utils.data.Dataset):
    """Weighted mixture of GPTDatasets (reference blended_dataset.py)."""

    def __init__(self, datasets: Sequence[torch.utils.data.Dataset],
                 weights: Sequence[float], num_samples: int):
        assert len(datasets) == len(weights) and datasets
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        h = _helpers()
        self.dataset_index = np.zeros(num_samples, dtype=np.int16)
        self.dataset_sample_index = np.zeros(num_samples, dtype=np.int64)
        h.build_blending_indices(self.dataset_index,
                                 self.dataset_sample_index, w,
                                 len(datasets), num_samples)
        self.datasets = list(datasets)
        self.num_samples = num_samples

    def __len__(self) -> int:
        return self.num_samples

    def __getitem__(self, i: int) -> torch.Tensor:
        d = int(self.dataset_index[i])
        s = int(self.dataset_sample_index[i])
        return self.datasets[d][s % len(self.datasets[d])]


def build_pretraining_dataset(data_paths: List[str], seq_length: int,
                              num_samples: int, seed: int = 1234):
    """data_paths: ["w1", "prefix1", "w2", "prefix2", ...] or ["prefix"]
    (reference megatron data_path convention)."""
    if len(data_paths) == 1:
        return GPTDataset(load_indexed_dataset(data_paths[0]), seq_length,
                          num_samples, seed)
    assert len(data_paths) % 2 == 0, \
        "data_path must be 'prefix' or 'w1 prefix1 w2 prefix2 ...'"
    weights = [float(data_paths[i]) for i in range(0, len(data_paths), 2)]
    prefixes = [data_paths[i] for i in range(1, len(data_paths), 2)]
    per = [max(int(num_samples * w / sum(weights)) + 1, 1) for w in weights]
    dsets = [GPTDataset(load_indexed_dataset(p), seq_length, n, seed + i)
             for i, (p, n) in enumerate(zip(prefixes, per))]
    return BlendedDataset(dsets, weights, num_samples)
