"""Tokenizer wrappers (Megatron-style API over offline vocab files).

Reference: core/runtime/datasets/megatron/megatron_tokenizer.py +
tokenizer.py (abstract tokenize/detokenize/vocab_size/eod over HF /
sentencepiece backends).  No network: callers supply the tokenizer file.
"""
from __future__ import annotations

from typing import List, Optional


class NullTokenizer:
    """Integer-passthrough tokenizer for synthetic pretraining data
    (ids are already ints; reference tokenizer.py NullTokenizer)."""

    def __init__(self, vocab_size: int):
        self._vocab = vocab_size

    @property
    def vocab_size(self) -> int:
        return self._vocab

    @property
    def eod(self) -> int:
        return self._vocab - 1

    def tokenize(self, text: str) -> List[int]:
        return [int(t) for t in text.split()]

    def detokenize(self, ids: List[int]) -> str:
        return " ".join(str(i) for i in ids)


class HFTokenizer:
    """tokenizers-library JSON file (e.g. tokenizer.json from an HF repo)."""

    def __init__(self, tokenizer_file: str):
        from tokenizers import Tokenizer
        self._t = Tokenizer.from_file(tokenizer_file)

    @property
    def vocab_size(self) -> int:
        return self._t.get_vocab_size()

    @property
    def eod(self) -> int:
        for tok in ("</s>", "<|endoftext|>", "<eos>"):
            i = self._t.token_to_id(tok)
            if i is not None:
                return i
        return self.vocab_size - 1

    def tokenize(self, text: str) -> List[int]:
        return self._t.encode(text).ids

    def detokenize(self, ids: List[int]) -> str:
        return self._t.decode(ids)


class SentencePieceTokenizer:
    """sentencepiece .model file (llama/t5 style)."""

    def __init__(self, model_file: str):
        import sentencepiece as spm
        self._sp = spm.SentencePieceProcessor(model_file=model_file)

    @property
    def vocab_size(self) -> int:
        return self._sp.vocab_size()

    @property
    def eod(self) -> int:
        return self._sp.eos_id() if self._sp.eos_id() >= 0 \
            else self.vocab_size - 1

    def tokenize(self, text: str) -> List[int]:
        return self._sp.encode(text)

    def detokenize(self, ids: List[int]) -> str:
        return self._sp.decode(ids)


def build_tokenizer(kind: str, file: Optional[str] = None,
                    vocab_size: int = 0):
    """kind: null | hf | sentencepiece."""
    if kind == "null":
        return NullTokenizer(vocab_size)
    if kind == "hf":
        return HFTokenizer(file)
    if kind == "sentencepiece":
        return SentencePieceTokenizer(file)
    raise ValueError(f"unknown tokenizer kind {kind!r}")
