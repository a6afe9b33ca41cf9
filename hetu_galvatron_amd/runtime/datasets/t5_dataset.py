"""T5 span-corruption pretraining dataset.

Reference role: the BERT/T5-style masked dataset builders the reference
vendors from Megatron (datasets/megatron + helpers.cpp
build_mapping_impl).  Re-derived from the T5 paper's objective: sample
token spans (mean length 3, noise density 15%), replace each span in the
encoder input with a sentinel id, and emit the decoder target as the
sentinel-delimited deleted spans.

Sentinels occupy the TOP of the vocab (ids vocab_size-1, vocab_size-2,
... like HF T5's <extra_id_k>).  Deterministic per (sample, seed).
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch


def corrupt_spans(tokens: np.ndarray, vocab_size: int, seed: int,
                  noise_density: float = 0.15, mean_span: float = 3.0,
                  max_sentinels: int = 100):
    """tokens [L] -> (enc_ids, dec_ids) numpy arrays (unpadded)."""
    rng = np.random.RandomState(seed)
    L = len(tokens)
    n_noise = max(1, int(round(L * noise_density)))
    n_spans = max(1, int(round(n_noise / mean_span)))
    n_spans = min(n_spans, max_sentinels, n_noise)
    # split n_noise into n_spans positive parts, and the remaining
    # L - n_noise tokens into n_spans + 1 (possibly empty) gaps
    def split(total, parts, min_v):
        cuts = np.sort(rng.choice(total - parts * min_v + parts - 1,
                                  parts - 1, replace=False)) \
            if parts > 1 else np.array([], dtype=int)
        sizes = np.diff(np.concatenate([[-1], cuts,
                                        [total - parts * min_v + parts - 1]]))
        return sizes - 1 + min_v

    span_sizes = split(n_noise, n_spans, 1)
    gap_sizes = split(L - n_noise + n_spans + 1, n_spans + 1, 1) - 1
    enc: List[int] = []
    dec: List[int] = []
    pos = 0
    for i in range(n_spans):
        g = int(gap_sizes[i])
        enc.extend(tokens[pos:pos + g])
        pos += g
        sentinel = vocab_size - 1 - i
        enc.append(sentinel)
        dec.append(sentinel)
        s = int(span_sizes[i])
        dec.extend(tokens[pos:pos + s])
        pos += s
    enc.extend(tokens[pos:])
    dec.append(vocab_size - 1 - n_spans)  # final sentinel = EOS marker
    return np.asarray(enc, dtype=np.int64), np.asarray(dec, dtype=np.int64)


class T5MaskedDataset(torch.utils.data.Dataset):
    """Wraps a token-sequence dataset (GPTDataset / synthetic) into
    (enc_input_ids, dec tokens [S_dec+1]) pairs for the enc-dec engine."""

    def __init__(self, inner: torch.utils.data.Dataset, enc_seq_len: int,
                 dec_seq_len: int, vocab_size: int, seed: int = 1234,
                 noise_density: float = 0.15, mean_span: float = 3.0):
        self.inner = inner
        self.enc_seq_len = enc_seq_len
        self.dec_seq_len = dec_seq_len
        self.vocab_size = vocab_size
        self.seed = seed
        self.noise_density = noise_density
        self.mean_span = mean_span

    def __len__(self) -> int:
        return len(self.inner)

    def __getitem__(self, i: int) -> Dict[str, torch.Tensor]:
        raw = self.inner[i]
        toks = raw.numpy() if torch.is_tensor(raw) else np.asarray(raw)
        # keep sentinel ids out of the data tokens
        toks = np.clip(toks, 0, self.vocab_size - 1 - 101)
        enc, dec = corrupt_spans(toks, self.vocab_size,
                                 self.seed * 100003 + i,
                                 self.noise_density, self.mean_span)

        def fit(a, n, pad):
            if len(a) >= n:
                return a[:n]
            return np.concatenate([a, np.full(n - len(a), pad,
                                              dtype=np.int64)])

        return {
            "enc_input_ids": torch.from_numpy(fit(enc, self.enc_seq_len, 0)),
            # decoder stream [S_dec + 1]: engine shifts input/labels
            "dec_tokens": torch.from_numpy(fit(dec, self.dec_seq_len + 1, 0)),
        }
