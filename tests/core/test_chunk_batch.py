"""Unit edges of the microbatch splitter (pipeline/engine.chunk_batch)."""
import torch

from hetu_galvatron_amd.runtime.pipeline.engine import chunk_batch


def _ctx(B, S=8):
    return {"batch_size": B, "seq_len": S,
            "input_ids": torch.arange(B * S).view(B, S),
            "labels": torch.arange(B * S).view(B, S)}


def test_even_split():
    mb = chunk_batch(_ctx(8), 4, dp=2)
    assert [m["batch_size"] for m in mb] == [2, 2, 2, 2]
    assert torch.equal(torch.cat([m["input_ids"] for m in mb]),
                       _ctx(8)["input_ids"])


def test_remainder_front_loaded():
    mb = chunk_batch(_ctx(10), 3, dp=2)  # 5 rows over 3 chunks -> 2,2,1
    assert [m["batch_size"] for m in mb] == [4, 4, 2]
    assert sum(m["batch_size"] for m in mb) == 10


def test_chunks_clamped_to_rows():
    mb = chunk_batch(_ctx(4), 8, dp=4)  # 1 row -> 1 chunk
    assert len(mb) == 1 and mb[0]["batch_size"] == 4


def test_dp1_odd_batch():
    mb = chunk_batch(_ctx(7), 3, dp=1)
    assert [m["batch_size"] for m in mb] == [3, 2, 2]
    assert torch.equal(torch.cat([m["labels"] for m in mb]),
                       _ctx(7)["labels"])
