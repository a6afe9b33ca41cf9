"""Profiler CPU tests: post-processing math on fabricated raw profiles,
hardware-bench script generation, runtime-profiler JSON plumbing."""
import json
import os

import pytest

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.profiler.hardware import HardwareProfiler
from hetu_galvatron_amd.profiler.model import ModelProfiler
from hetu_galvatron_amd.search.engine import read_computation_profile


def test_computation_differencing():
    # fabricate: per-layer 4.4 ms/bsz + other 2.0 ms + 0.5 const
    raw = {}
    for ln in (2, 4):
        for b in (1, 2, 4, 8):
            raw[f"layernum[{ln}]_bsz{b}_seq4096"] = ln * (4.4 * b + 0.5) + 2.0 * b
    m, c = read_computation_profile(raw, 4096)
    assert abs(m - 4.4) < 1e-6 and abs(c - 0.5) < 1e-6


def test_memory_postprocess(tmp_path):
    cfg = load_config(base={"model": {"model_name": "tiny-llama"},
                            "profile": {"profile_dir": str(tmp_path)}})
    mp = ModelProfiler(cfg)
    param, other_p = 840.0, 2100.0
    act = {1: 520.0, 2: 260.0, 4: 130.0, 8: 65.0}
    other_act = 600.0
    bsz = 8
    raw = {}
    for ln in (1, 2):
        for tp in (1, 2, 4, 8):
            k = f"1_{tp}_{8 // tp}/layernum[{ln}]_bsz{bsz}_seq4096_rank0"
            raw[k + "_ms"] = 4 * (ln * param / tp + other_p)
            raw[k + "_act"] = ln * act[tp] * bsz + other_act * bsz
            raw[k + "_act_peak"] = raw[k + "_act"] * 1.2
        k = f"1_1_8_c/layernum[{ln}]_bsz{bsz}_seq4096_rank0"
        raw[k + "_ms"] = 4 * (ln * param + other_p)
        raw[k + "_act"] = ln * 34.0 * bsz + other_act * bsz
        raw[k + "_act_peak"] = raw[k + "_act"]
    parsed = mp.process_memory_data(raw, write=True)
    lt = parsed["layertype_0"]
    assert abs(lt["parameter_size"] - param) < 1e-6
    assert abs(lt["tp_activation_per_bsz_dict"]["1"] - act[1]) < 1e-6
    assert abs(lt["tp_activation_per_bsz_dict"]["4"] - act[4]) < 1e-6
    assert abs(lt["tp_activation_per_bsz_dict"]["checkpoint"] - 34.0) < 1e-6
    assert abs(parsed["other"]["parameter_size"] - other_p) < 1e-6
    assert abs(parsed["other"]["tp_activation_per_bsz_dict"]["1"] - other_act) < 1e-6
    assert os.path.exists(os.path.join(str(tmp_path),
                                       "model_profile_bf16_tiny-llama.json"))


def test_hardware_script_generation(tmp_path):
    cfg = load_config(base={"model": {"model_name": "tiny-llama"}})
    hp = HardwareProfiler(cfg)
    paths = hp.generate_scripts(str(tmp_path))
    assert len(paths) == 4
    for p in paths:
        body = open(p).read()
        assert "torch.distributed.run" in body
        assert "hardware_bench" in body
        assert "--master-addr 127.0.0.1" in body


def test_runtime_profiler_cpu_noop(tmp_path):
    from hetu_galvatron_amd.profiler.runtime import RuntimeProfiler
    rp = RuntimeProfiler(enabled=False)
    rp.profile_memory("Before-Fwd")
    rp.time_start()
    assert rp.time_end() is None
    rp.save_time_profile(os.path.join(tmp_path, "t.json"), "layernum[2]_bsz1_seq128")
    d = json.load(open(os.path.join(tmp_path, "t.json")))
    assert "layernum[2]_bsz1_seq128" in d


def test_t5_model_profiler_two_axis():
    """Fabricated (enc,dec) sweeps -> layertype_0/1 with per-type fct;
    the parsed profile loads into the multi-layer-type search."""
    import json
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.profiler.model import T5ModelProfiler

    cfg = load_config(base={
        "model": {"model_name": "tiny-t5"},
        "profile": {"profile_min_layer_num": 1,
                    "profile_max_layer_num": 2,
                    "profile_fixed_batch_size": 4}})
    prof = T5ModelProfiler(cfg)
    seq = cfg.model.seq_length
    # enc layer: 2.0 ms/layer/bsz + 0.1; dec: 3.0/bsz + 0.2; base 1.0
    comp = {}
    for enc, dec in ((1, 1), (2, 1), (1, 2)):
        for b in (1, 2, 4):
            comp[f"layernum[{enc},{dec}]_bsz{b}_seq{seq}"] = \
                1.0 + enc * (2.0 * b + 0.1) + dec * (3.0 * b + 0.2)
    mem = {}
    for enc, dec in ((1, 1), (2, 1), (1, 2)):
        mem[f"1_1_1/layernum[{enc},{dec}]_bsz4_seq{seq}_rank0_ms"] = \
            100.0 + enc * 40.0 + dec * 60.0
        mem[f"1_1_1/layernum[{enc},{dec}]_bsz4_seq{seq}_rank0_act"] = \
            20.0 + enc * 8.0 + dec * 12.0
        mem[f"1_1_1/layernum[{enc},{dec}]_bsz4_seq{seq}_rank0_act_peak"] = 0.0
    parsed = prof.process_t5_data(comp, mem, write=False)
    lt0, lt1 = parsed["layertype_0"], parsed["layertype_1"]
    assert abs(lt0["fct_linear"][0] - 2.0) < 1e-6
    assert abs(lt1["fct_linear"][0] - 3.0) < 1e-6
    assert abs(lt0["parameter_size"] - 10.0) < 1e-6   # 40/4
    assert abs(lt1["parameter_size"] - 15.0) < 1e-6
    assert abs(lt0["tp_activation_per_bsz_dict"]["1"] - 2.0) < 1e-6


def test_record_function_scopes():
    """galvatron:: tracing scopes appear in a torch.profiler capture."""
    import torch
    from torch.profiler import ProfilerActivity, profile
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    it = get_train_iterator(cfg, torch.device("cpu"))
    with profile(activities=[ProfilerActivity.CPU]) as prof:
        model.forward_backward(next(it))
    names = {e.key for e in prof.key_averages()}
    assert any("galvatron::grad_sync_start" in n for n in names), names


def test_train_cli_profile_mode_t5_keys(tmp_path):
    """profile.profile=1 computation run writes t5 'layernum[enc,dec]'
    keys (consumed by T5ModelProfiler.process_t5_data)."""
    import json
    import os
    from hetu_galvatron_amd.cli.train import main
    main(["model.model_name=tiny-t5",
          "parallel.mixed_precision=fp32",
          "model.num_hidden_layers=1", "model.num_decoder_layers=1",
          "train.global_train_batch_size=2", "train.train_iters=2",
          "train.lr=1e-4", "train.lr_decay_style=constant",
          "train.distributed_backend=gloo",
          "profile.profile=1", "profile.profile_type=computation",
          f"profile.profile_dir={tmp_path}"])
    path = os.path.join(tmp_path, "computation_profiling_fp32_tiny-t5.json")
    d = json.load(open(path))
    assert any(k.startswith("layernum[1,1]_bsz2_seq") for k in d), d


def _sp_time_worker(rank, world):
    import torch.distributed as dist
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank, world_size=world)
    import argparse
    from hetu_galvatron_amd.profiler.hardware_bench import bench_sp_time
    args = argparse.Namespace(start_mb=1, end_mb=2, warmup_iters=1,
                              measure_iters=2)
    out = bench_sp_time(args, rank, world)
    dist.barrier()
    return out


def test_sp_time_worker_per_size_keys():
    """sp_time sweeps per sub-group size (cost-model input curves)."""
    from tests.utils import run_distributed
    res = run_distributed(_sp_time_worker, world_size=2)
    keys = set(res[0])
    assert "allreduce_size_2_1MB_time" in keys
    assert "allreduce_size_2_2MB_time" in keys


def _hw_worker(rank, world, op):
    import torch.distributed as dist
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank, world_size=world)
    import argparse
    from hetu_galvatron_amd.profiler import hardware_bench as hb
    args = argparse.Namespace(message_mb=1, warmup_iters=1, measure_iters=2,
                              start_mb=1, end_mb=1)
    out = {"allreduce": hb.bench_allreduce,
           "p2p": hb.bench_p2p}[op](args, rank, world)
    dist.barrier()
    return out


def test_allreduce_worker_keys():
    from tests.utils import run_distributed
    res = run_distributed(_hw_worker, world_size=2, args=("allreduce",))
    assert "allreduce_size_2_consec_1" in res[0]
    assert res[0]["allreduce_size_2_consec_1"] > 0


def test_p2p_worker_keys():
    from tests.utils import run_distributed
    res = run_distributed(_hw_worker, world_size=2, args=("p2p",))
    assert any(k.startswith("pp_size_") for k in res[0])


def _overlap_worker(rank, world):
    import torch.distributed as dist
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank, world_size=world)
    import argparse
    from hetu_galvatron_amd.profiler.hardware_bench import bench_overlap
    out = bench_overlap(argparse.Namespace(), rank, world)
    dist.barrier()
    return out


def test_overlap_worker_cpu_default():
    from tests.utils import run_distributed
    res = run_distributed(_overlap_worker, world_size=2)
    assert res[0]["overlap_coe"] >= 1.0


def test_profile_hardware_cli_scripts_only(tmp_path, monkeypatch):
    # CLI entry (reference profile_hardware.py:7): --scripts-only writes the
    # torchrun sweep scripts without launching anything
    monkeypatch.chdir(tmp_path)
    from hetu_galvatron_amd.cli.profile_hardware import main
    main(["--scripts-only",
          "profile_hardware.num_nodes=1",
          "profile_hardware.num_gpus_per_node=8",
          f"profile_hardware.hardware_config_dir={tmp_path}/hw"])
    import glob
    scripts = glob.glob(str(tmp_path / "scripts" / "*.sh"))
    assert len(scripts) >= 4, scripts
    names = " ".join(scripts)
    for op in ("allreduce", "p2p", "sp_time", "overlap"):
        assert op in names, f"missing sweep script for {op}"
    # each script must be a single-node torchrun invocation on loopback
    body = open(scripts[0]).read()
    assert "torchrun" in body or "torch.distributed.run" in body
    assert "127.0.0.1" in body
