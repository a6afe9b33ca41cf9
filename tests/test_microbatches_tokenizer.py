"""Batch ramp-up calculator + tokenizer wrappers (reference:
num_microbatches_calculator.py, megatron_tokenizer.py)."""
import pytest

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.runtime.datasets.tokenizer import (NullTokenizer,
                                                           build_tokenizer)
from hetu_galvatron_amd.runtime.optimizer.microbatches import (
    ConstantBatchCalculator, RampupBatchCalculator, build_batch_calculator)


def test_constant_calculator():
    c = ConstantBatchCalculator(64, 2, 8)
    c.update(10_000)
    assert c.get() == (64, 4)


def test_rampup_calculator():
    r = RampupBatchCalculator(start=16, increment=16, ramp_samples=300,
                              global_batch_size=64, micro_batch_size=2, dp=8)
    assert r.get() == (16, 1)
    r.update(0)
    assert r.get() == (16, 1)
    r.update(150)                   # 1 of 3 steps passed
    assert r.get()[0] == 32
    r.update(299)
    assert r.get()[0] == 48
    r.update(300)
    assert r.get() == (64, 4)
    r.update(10**9)
    assert r.get() == (64, 4)


def test_rampup_validation():
    with pytest.raises(AssertionError):
        RampupBatchCalculator(10, 16, 100, 64, 2, 8)  # start not divisible
    with pytest.raises(AssertionError):
        RampupBatchCalculator(16, 12, 100, 64, 2, 8)  # (64-16)%12 != 0


def test_build_from_config():
    cfg = load_config(base={"model": {"model_name": "tiny-llama"},
                            "train": {"global_train_batch_size": 32,
                                      "rampup_batch_size": "8,8,100"}})
    calc = build_batch_calculator(cfg, dp=2, micro_batch_size=4)
    assert isinstance(calc, RampupBatchCalculator)
    assert calc.get() == (8, 1)
    cfg2 = load_config(base={"model": {"model_name": "tiny-llama"}})
    assert isinstance(build_batch_calculator(cfg2, 1, 8),
                      ConstantBatchCalculator)


def test_null_tokenizer_roundtrip():
    t = NullTokenizer(512)
    assert t.vocab_size == 512 and t.eod == 511
    assert t.detokenize(t.tokenize("1 2 3")) == "1 2 3"
    assert isinstance(build_tokenizer("null", vocab_size=16), NullTokenizer)


def test_hf_tokenizer_if_available():
    try:
        import tokenizers
    except ImportError:
        pytest.skip("tokenizers not installed")
    from tokenizers import Tokenizer, models
    import tempfile, os
    tok = Tokenizer(models.WordLevel({"hello": 0, "world": 1, "</s>": 2},
                                     unk_token=None))
    from tokenizers.pre_tokenizers import Whitespace
    tok.pre_tokenizer = Whitespace()
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "tok.json")
        tok.save(p)
        t = build_tokenizer("hf", p)
        assert t.vocab_size == 3 and t.eod == 2
        assert t.tokenize("hello world") == [0, 1]


def test_train_cli_with_rampup(tmp_path):
    """End-to-end: train CLI ramps the global batch without error."""
    from hetu_galvatron_amd.cli.train import main
    main(["model.model_name=tiny-llama",
          "parallel.mixed_precision=fp32",
          "train.global_train_batch_size=8",
          "train.rampup_batch_size=2,2,12",
          "train.train_iters=6", "train.lr=1e-4",
          "train.lr_decay_style=constant",
          "train.distributed_backend=gloo"])


def test_train_cli_t5(tmp_path):
    """t5 through the train CLI (enc-dec batches, relative bias)."""
    from hetu_galvatron_amd.cli.train import main
    main(["model.model_name=tiny-t5",
          "parallel.mixed_precision=fp32",
          "train.global_train_batch_size=2", "train.train_iters=2",
          "train.lr=1e-4", "train.lr_decay_style=constant",
          "train.distributed_backend=gloo"])


def test_torchrun_train_moe_pp2(tmp_path):
    """cli.train end-to-end with MoE across a pipeline split: the aux
    tracker's metric reduction must handle per-stage key divergence
    (regression for the dict-gather fix)."""
    import socket
    import subprocess
    import sys

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes", "1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port),
           "-m", "hetu_galvatron_amd.cli.train",
           "model.model_name=tiny-moe",
           "parallel.pp_deg=2", "parallel.chunks=2",
           "parallel.mixed_precision=fp32",
           "train.global_train_batch_size=4", "train.train_iters=2",
           "train.lr=1e-3", "train.lr_decay_style=constant",
           "train.distributed_backend=gloo",
           f"logging.tensorboard_dir={tmp_path}"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "iter     2" in r.stdout
    import json
    lines = [json.loads(l) for l in
             open(tmp_path / "metrics.jsonl").read().splitlines()]
    moe_keys = [k for l in lines for k in l if k.startswith("moe/")]
    assert moe_keys, "aux-loss metrics missing from the pp2 MoE run"
