"""Data pipeline: synthetic causal-LM batches (+ batch-context assembly).

Reference: galvatron/core/runtime/dataloader.py:36-567 (FakeCausalLMDataset,
get_batch, loss averaging).  The megatron mmap dataset stack hooks in via
data.dataset="megatron" (see datasets/).

Design note: every rank materializes the FULL global batch of token ids
(ids are ~2 bytes/token of int64 -> negligible vs activations) and each
layer module slices batch/sequence by its own layout — this is what makes
per-layer dp/tp/cp degrees composable without a per-layer dataloader.
"""
from __future__ import annotations

from typing import Dict, Iterator, Optional

import torch

from ..config import GalvatronConfig


class SyntheticCausalLMDataset(torch.utils.data.Dataset):
    """Deterministic synthetic token stream (same on every rank for a given
    seed + index; reference: dataloader.py:36 FakeCausalLMDataset)."""

    def __init__(self, vocab_size: int, seq_length: int, size: int = 1024,
                 seed: int = 1234):
        self.vocab_size = vocab_size
        self.seq_length = seq_length
        self.size = size
        self.seed = seed

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, idx: int) -> torch.Tensor:
        g = torch.Generator()
        g.manual_seed(self.seed * 100003 + idx)
        return torch.randint(0, self.vocab_size, (self.seq_length + 1,),
                             generator=g)


def build_batch_context(tokens: torch.Tensor, device,
                        eod_token: int = None,
                        eod_mask_loss: bool = False) -> Dict:
    """tokens: [B, S+1] -> ctx with input_ids/labels [B, S].
    eod_mask_loss (reference dataloader get_batch loss_mask): zero the
    loss on end-of-document labels; loss_denom is the GLOBAL unmasked
    token count (identical on every rank -> grads normalize exactly)."""
    tokens = tokens.to(device)
    ctx = {
        "input_ids": tokens[:, :-1].contiguous(),
        "labels": tokens[:, 1:].contiguous(),
        "batch_size": tokens.shape[0],
        "seq_len": tokens.shape[1] - 1,
    }
    if eod_mask_loss:
        assert eod_token is not None, "eod_mask_loss needs data.eod_token_id"
        mask = (ctx["labels"] != eod_token).to(torch.float32)
        ctx["loss_mask"] = mask
        ctx["loss_denom"] = float(mask.sum())
    return ctx


def build_enc_dec_batch_context(enc_ids: torch.Tensor,
                                dec_tokens: torch.Tensor, device) -> Dict:
    """t5: enc_ids [B, S_enc]; dec_tokens [B, S_dec+1]."""
    ctx = build_batch_context(dec_tokens, device)
    ctx["enc_input_ids"] = enc_ids.to(device).contiguous()
    ctx["enc_seq_len"] = enc_ids.shape[1]
    return ctx


def split_range(n: int, split: str, which: str) -> tuple:
    """[lo, hi) index range of the train/valid/test partition under the
    reference's weight string (data.split, e.g. "969,30,1")."""
    w = [float(x) for x in split.split(",")]
    while len(w) < 3:
        w.append(0.0)
    tot = sum(w) or 1.0
    n_train = int(n * w[0] / tot)
    n_valid = int(n * w[1] / tot)
    lo, hi = {"train": (0, n_train),
              "valid": (n_train, n_train + n_valid),
              "test": (n_train + n_valid, n)}[which]
    if hi <= lo:  # degenerate split: fall back to the whole set
        return 0, n
    return lo, hi


def get_train_iterator(cfg: GalvatronConfig, device,
                       global_batch: Optional[int] = None,
                       split: str = "train") -> Iterator[Dict]:
    """Yield batch contexts of the global batch size, cycling the dataset
    (reference: dataloader.py:462 get_train_valid_test_data_iterators).
    split selects the train/valid/test partition per data.split."""
    B = global_batch or cfg.train.global_train_batch_size
    if cfg.data.dataset == "megatron" and cfg.data.data_path:
        from .datasets import build_pretraining_dataset
        ds = build_pretraining_dataset(
            cfg.data.data_path, cfg.model.seq_length,
            num_samples=max(B * max(cfg.train.train_iters, 1),
                            cfg.data.synthetic_dataset_size),
            seed=cfg.train.seed)
    else:
        ds = SyntheticCausalLMDataset(
            cfg.model.vocab_size, cfg.model.seq_length,
            size=max(cfg.data.synthetic_dataset_size, B), seed=cfg.train.seed)
    if cfg.model.model_type == "t5":
        s_enc = cfg.model.encoder_seq_length or cfg.model.seq_length
        if cfg.data.dataset == "megatron" and cfg.data.data_path:
            # span corruption over the token stream (t5 pretraining
            # objective; datasets/t5_dataset.py)
            from .datasets.t5_dataset import T5MaskedDataset
            t5ds = T5MaskedDataset(ds, s_enc, cfg.model.seq_length,
                                   cfg.model.vocab_size, seed=cfg.train.seed)
            idx = 0
            while True:
                items = [t5ds[(idx + i) % len(t5ds)] for i in range(B)]
                idx = (idx + B) % len(t5ds)
                enc = torch.stack([it["enc_input_ids"] for it in items])
                dec = torch.stack([it["dec_tokens"] for it in items])
                yield build_enc_dec_batch_context(enc, dec, device)
        g = torch.Generator().manual_seed(cfg.train.seed + 77)
        idx = 0
        while True:
            batch = torch.stack([ds[(idx + i) % len(ds)] for i in range(B)])
            idx = (idx + B) % len(ds)
            enc = torch.randint(0, cfg.model.vocab_size, (B, s_enc),
                                generator=g)
            yield build_enc_dec_batch_context(enc, batch, device)
    lo, hi = split_range(len(ds), cfg.data.split, split)
    n = hi - lo
    idx = 0
    while True:
        batch = torch.stack([ds[lo + (idx + i) % n] for i in range(B)])
        idx = (idx + B) % n
        yield build_batch_context(batch, device,
                                  eod_token=cfg.data.eod_token_id,
                                  eod_mask_loss=cfg.data.eod_mask_loss)
