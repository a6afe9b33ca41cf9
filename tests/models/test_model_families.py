"""Per-family single-process training smokes + loss-decrease sanity
(reference: tests/models/test_model_correctness.py)."""
import pytest
import torch

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.runtime import (
    GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)


@pytest.mark.parametrize("name", ["tiny-llama", "tiny-gpt", "tiny-moe"])
def test_family_trains(name):
    cfg = load_config(base={
        "model": {"model_name": name},
        "train": {"global_train_batch_size": 4, "train_iters": 8,
                  "lr": 5e-3, "lr_decay_style": "constant"},
    })
    torch.manual_seed(0)
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    m = GalvatronModel(cfg, device=device)
    opt, sched = get_optimizer_and_param_scheduler(m.stage_model, cfg)
    it = get_train_iterator(cfg, device)
    batch = next(it)  # overfit one batch: loss must drop
    losses = []
    for _ in range(8):
        opt.zero_grad()
        st = m.forward_backward(batch)
        opt.step()
        sched.step()
        losses.append(st.loss)
    assert losses[-1] < losses[0] - 0.05, losses


def test_presets_resolve():
    from hetu_galvatron_amd.config.model_configs import MODEL_PRESETS
    for name in MODEL_PRESETS:
        cfg = load_config(base={"model": {"model_name": name}})
        assert cfg.model.hidden_size > 0
        if not cfg.model.kv_channels:
            assert cfg.model.head_dim * cfg.model.num_attention_heads \
                == cfg.model.hidden_size
