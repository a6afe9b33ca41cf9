"""The driver runs `bench.py` at N=1,2,4,8; its auto plan search must
succeed on every N with the committed measured profiles.  This pins the
CPU-visible part of that path (the search itself — no GPU needed)."""
import pytest

from hetu_galvatron_amd.config import load_config


class _Args:
    model = "llama-3-8b"
    seq_len = 4096
    batch_per_gpu = 8
    chunks = 0
    pp = 1
    tp = 1
    dp_type = "zero2"
    checkpoint = False
    plan = "auto"


@pytest.mark.parametrize("world", [1, 2, 4, 8])
def test_bench_plan_search_succeeds(world):
    import bench
    args = _Args()
    cfg = load_config(base={
        "model": {"model_name": args.model, "seq_length": args.seq_len},
        "train": {"global_train_batch_size": args.batch_per_gpu * world,
                  "train_iters": 8, "lr": 1e-4,
                  "lr_decay_style": "constant",
                  "distributed_backend": "nccl"},
        "parallel": {"pp_deg": 1, "global_tp_deg": 1,
                     "default_dp_type": "zero2", "chunks": 2,
                     "mixed_precision": "bf16"},
    })
    plan = bench.search_bench_plan(cfg, world, args)
    assert plan is not None, f"bench auto-search found no plan at N={world}"
    assert plan.global_bsz == args.batch_per_gpu * world
    # one full replica per GPU fits in 288 GB: search must not burn the
    # budget on tp/pp/ckpt at bench shapes
    assert plan.pp_deg == 1
    assert all(t == 1 for t in plan.tp_sizes_enc)
    assert not any(plan.checkpoint_flags)
    desc = bench.plan_desc(plan, world)
    assert desc.startswith("searched:")
