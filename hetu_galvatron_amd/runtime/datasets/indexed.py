"""Memory-mapped indexed token dataset (.bin tokens + .idx offsets).

Reference behavior: galvatron/core/runtime/datasets/megatron/
indexed_dataset.py (mmap .bin/.idx document store).  Own, simpler format:

  <prefix>.bin : raw tokens, little-endian, dtype from the .idx header
  <prefix>.idx : header {magic 'GALVIDX1', dtype code, n_docs} +
                 int64 doc offsets [n_docs + 1] (token units)
"""
from __future__ import annotations

import struct
from typing import List, Sequence

import numpy as np

MAGIC = b"GALVIDX1"
DTYPES = {1: np.uint16, 2: np.int32, 3: np.int64}
DTYPE_CODES = {np.dtype(v).name: k for k, v in DTYPES.items()}


class IndexedDatasetBuilder:
    def __init__(self, prefix: str, dtype=np.int32):
        self.prefix = prefix
        self.dtype = np.dtype(dtype)
        self._bin = open(prefix + ".bin", "wb")
        self.offsets: List[int] = [0]

    def add_document(self, tokens: Sequence[int]) -> None:
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.offsets.append(self.offsets[-1] + len(arr))

    def finalize(self) -> None:
        self._bin.close()
        with open(self.prefix + ".idx", "wb") as f:
            f.write(MAGIC)
            f.write(struct.pack("<BQ", DTYPE_CODES[self.dtype.name],
                                len(self.offsets) - 1))
            f.write(np.asarray(self.offsets, dtype=np.int64).tobytes())


class IndexedDataset:
    def __init__(self, prefix: str):
        with open(prefix + ".idx", "rb") as f:
            magic = f.read(8)
            assert magic == MAGIC, f"bad index file {prefix}.idx"
            code, n_docs = struct.unpack("<BQ", f.read(9))
            self.offsets = np.frombuffer(f.read(8 * (n_docs + 1)),
                                         dtype=np.int64)
        self.dtype = DTYPES[code]
        self.tokens = np.memmap(prefix + ".bin", dtype=self.dtype, mode="r")
        self.n_docs = n_docs

    def __len__(self) -> int:
        return self.n_docs

    @property
    def doc_lens(self) -> np.ndarray:
        return (self.offsets[1:] - self.offsets[:-1]).astype(np.int64)

    def doc(self, i: int) -> np.ndarray:
        return self.tokens[self.offsets[i]:self.offsets[i + 1]]

    def read_span(self, doc: int, offset: int, length: int) -> np.ndarray:
        """Read `length` tokens starting at (doc, offset), wrapping docs
        (and the corpus) as needed."""
        out = np.empty(length, dtype=np.int64)
        filled = 0
        d, off = doc, offset
        while filled < length:
            chunk = self.tokens[self.offsets[d] + off:self.offsets[d + 1]]
            take = min(len(chunk), length - filled)
            out[filled:filled + take] = chunk[:take]
            filled += take
            d = (d + 1) % self.n_docs
            off = 0
        return out
