"""MetricsLogger sinks (JSONL always; tb/wandb degrade gracefully)."""
import json

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.utils.logging import MetricsLogger


def test_jsonl_sink(tmp_path):
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "logging": {"tensorboard_dir": str(tmp_path)}})
    ml = MetricsLogger(cfg, rank=0)
    ml.log({"loss": 1.5, "lr": 1e-4}, step=0)
    ml.log({"loss": 1.2, "lr": 1e-4}, step=1)
    ml.close()
    lines = [json.loads(l) for l in
             open(tmp_path / "metrics.jsonl").read().splitlines()]
    assert [l["step"] for l in lines] == [0, 1]
    assert abs(lines[1]["loss"] - 1.2) < 1e-9


def test_nonzero_rank_disabled(tmp_path):
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "logging": {"tensorboard_dir": str(tmp_path)}})
    ml = MetricsLogger(cfg, rank=1)
    ml.log({"loss": 1.0}, 0)
    ml.close()
    assert not (tmp_path / "metrics.jsonl").exists()


def test_global_memory_buffer_reuse():
    import torch
    from hetu_galvatron_amd.core.parallel_state import GlobalMemoryBuffer
    b = GlobalMemoryBuffer()
    t1 = b.get_tensor((4, 8), torch.float32, "ws")
    ptr1 = t1.data_ptr()
    t2 = b.get_tensor((2, 8), torch.float32, "ws")   # smaller: same storage
    assert t2.data_ptr() == ptr1 and t2.shape == (2, 8)
    t3 = b.get_tensor((16, 8), torch.float32, "ws")  # larger: grows
    assert t3.shape == (16, 8)
    t4 = b.get_tensor((4, 4), torch.float16, "other")
    assert t4.dtype == torch.float16
