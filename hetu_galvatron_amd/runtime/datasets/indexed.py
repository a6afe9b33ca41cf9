"""Memory-mapped indexed token dataset (.bin tokens + .idx offsets).

Reference behavior: galvatron/core/runtime/datasets/megatron/
indexed_dataset.py (mmap .bin/.idx document store).  Two on-disk formats:

  * native (`GALVIDX1`): header {magic, dtype code, n_docs} + int64 doc
    offsets [n_docs + 1] in token units — what IndexedDatasetBuilder writes.
  * Megatron `MMIDIDX` (indexed_dataset.py:38-205): header {magic
    'MMIDIDX\\x00\\x00', version u64=1, dtype code u8}, sequence_count u64,
    document_count u64, sizes i32[count], byte pointers i64[count],
    doc_idx i64[doc_count] — so corpora preprocessed with Megatron's
    preprocess_data.py drop in unchanged.

`load_indexed_dataset(prefix)` sniffs the magic and returns either reader;
both expose the same doc()/read_span()/doc_lens interface.
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


MEGATRON_MAGIC = b"MMIDIDX\x00\x00"
# Megatron DType enum (indexed_dataset.py:40-50)
MEGATRON_DTYPES = {1: np.uint8, 2: np.int8, 3: np.int16, 4: np.int32,
                   5: np.int64, 6: np.float64, 7: np.float32, 8: np.uint16}
MEGATRON_DTYPE_CODES = {np.dtype(v).name: k for k, v in MEGATRON_DTYPES.items()}


class MegatronIndexedDataset:
    """Reader for standard Megatron-preprocessed `.bin`/`.idx` corpora
    (MMapIndexedDataset layout, reference indexed_dataset.py:233-376).
    Sequences are treated as documents (Megatron's GPTDataset does the
    same via document_indices); exposes the native reader's interface."""

    def __init__(self, prefix: str):
        with open(prefix + ".idx", "rb") as f:
            magic = f.read(9)
            assert magic == MEGATRON_MAGIC, f"bad Megatron index {prefix}.idx"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1, f"unsupported MMIDIDX version {version}"
            (code,) = struct.unpack("<B", f.read(1))
            count, doc_count = struct.unpack("<QQ", f.read(16))
            self.sizes = np.frombuffer(f.read(4 * count), dtype=np.int32)
            self.pointers = np.frombuffer(f.read(8 * count), dtype=np.int64)
            self.doc_idx = np.frombuffer(f.read(8 * doc_count),
                                         dtype=np.int64)
        self.dtype = MEGATRON_DTYPES[code]
        itemsize = np.dtype(self.dtype).itemsize
        self.tokens = np.memmap(prefix + ".bin", dtype=self.dtype, mode="r")
        # token-unit offsets [count + 1]: pointers are byte offsets; the
        # data is contiguous, so offsets derive from pointers/itemsize with
        # the final boundary from the last size
        offs = np.empty(count + 1, dtype=np.int64)
        offs[:count] = self.pointers // itemsize
        offs[count] = (count and offs[count - 1] + self.sizes[count - 1])
        self.offsets = offs
        self.n_docs = int(count)

    def __len__(self) -> int:
        return self.n_docs

    @property
    def doc_lens(self) -> np.ndarray:
        return self.sizes.astype(np.int64)

    doc = IndexedDataset.doc
    read_span = IndexedDataset.read_span


class MegatronIndexedDatasetBuilder:
    """Writer producing the standard MMIDIDX layout (for round-trip tests
    and for exporting native corpora to Megatron tooling)."""

    def __init__(self, prefix: str, dtype=np.int32):
        self.prefix = prefix
        self.dtype = np.dtype(dtype)
        self._bin = open(prefix + ".bin", "wb")
        self.sizes: List[int] = []

    def add_document(self, tokens: Sequence[int]) -> None:
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.sizes.append(len(arr))

    def finalize(self) -> None:
        self._bin.close()
        count = len(self.sizes)
        pointers = np.zeros(count, dtype=np.int64)
        if count > 1:
            np.cumsum(np.asarray(self.sizes[:-1], dtype=np.int64) *
                      self.dtype.itemsize, out=pointers[1:])
        with open(self.prefix + ".idx", "wb") as f:
            f.write(MEGATRON_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", MEGATRON_DTYPE_CODES[self.dtype.name]))
            f.write(struct.pack("<QQ", count, count))
            f.write(np.asarray(self.sizes, dtype=np.int32).tobytes())
            f.write(pointers.tobytes())
            f.write(np.arange(count, dtype=np.int64).tobytes())


def load_indexed_dataset(prefix: str):
    """Open `prefix.idx`/`prefix.bin`, sniffing the format magic: native
    GALVIDX1 or Megatron MMIDIDX."""
    with open(prefix + ".idx", "rb") as f:
        head = f.read(9)
    if head.startswith(MAGIC):
        return IndexedDataset(prefix)
    if head == MEGATRON_MAGIC:
        return MegatronIndexedDataset(prefix)
    raise ValueError(f"{prefix}.idx: unknown index magic {head!r}")
