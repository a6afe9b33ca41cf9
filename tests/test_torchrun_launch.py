"""End-to-end torchrun launch of the train CLI (the exact launcher the
driver uses for bench.py): 2 ranks, gloo, tiny model."""
import socket
import subprocess
import sys

import pytest


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.distributed
@pytest.mark.slow
def test_torchrun_train_cli(tmp_path):
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes", "1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()),
           "-m", "hetu_galvatron_amd.cli.train",
           "model.model_name=tiny-llama",
           "parallel.mixed_precision=fp32",
           "train.global_train_batch_size=4", "train.train_iters=2",
           "train.lr=1e-3", "train.lr_decay_style=constant",
           "train.distributed_backend=gloo",
           f"logging.tensorboard_dir={tmp_path}"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "iter     2" in r.stdout
    assert (tmp_path / "metrics.jsonl").exists()
