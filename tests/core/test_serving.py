"""Continuous-batching engine: dynamic join/leave produces EXACTLY the
tokens each request would get standalone (scheduling must not change
numerics; fp32 CPU)."""
import torch

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.runtime import GalvatronModel
from hetu_galvatron_amd.runtime.inference import GalvatronGenerator
from hetu_galvatron_amd.runtime.serving import ContinuousBatchingEngine


def make_model():
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    torch.manual_seed(0)
    return GalvatronModel(cfg)


def test_continuous_batching_matches_standalone():
    model = make_model()
    gen = GalvatronGenerator(model, max_batch=1, max_seq=64)
    eng = ContinuousBatchingEngine(model, max_slots=3, max_seq=64)
    torch.manual_seed(5)
    prompts = [torch.randint(0, 512, (n,)) for n in (5, 9, 3, 7)]
    budgets = [4, 6, 2, 3]
    want = [gen.generate(p.unsqueeze(0), max_new_tokens=b,
                         temperature=0.0)[0, p.shape[0]:].tolist()
            for p, b in zip(prompts, budgets)]

    # requests join at different steps; request 3 joins after 0 leaves
    r0 = eng.add_request(prompts[0], budgets[0])
    r1 = eng.add_request(prompts[1], budgets[1])
    eng.step()
    r2 = eng.add_request(prompts[2], budgets[2])
    steps = 0
    r3 = None
    while eng.n_active or r3 is None:
        eng.step()
        steps += 1
        if r3 is None and eng.free and steps >= 3:
            r3 = eng.add_request(prompts[3], budgets[3])
        assert steps < 50
    got = [eng.outputs[r] for r in (r0, r1, r2, r3)]
    assert got == want, (got, want)


def test_slot_reuse_and_pool_limit():
    model = make_model()
    eng = ContinuousBatchingEngine(model, max_slots=1, max_seq=64)
    r0 = eng.add_request(torch.randint(0, 512, (4,)), 1)  # done at prefill
    assert eng.n_active == 0 and len(eng.outputs[r0]) == 1
    r1 = eng.add_request(torch.randint(0, 512, (4,)), 2)
    assert eng.n_active == 1
    eng.step()
    assert eng.n_active == 0 and len(eng.outputs[r1]) == 2


def test_http_serve_endpoint():
    """FastAPI /generate over the continuous-batching engine (TestClient;
    concurrent requests share decode steps)."""
    import threading
    from fastapi.testclient import TestClient
    from hetu_galvatron_amd.cli.serve import build_app

    model = make_model()
    eng = ContinuousBatchingEngine(model, max_slots=2, max_seq=64)
    app = build_app(eng, threading.Lock())
    gen = GalvatronGenerator(model, max_batch=1, max_seq=64)
    torch.manual_seed(8)
    prompt = torch.randint(0, 512, (6,))
    want = gen.generate(prompt.unsqueeze(0), max_new_tokens=4,
                        temperature=0.0)[0, 6:].tolist()
    with TestClient(app) as client:
        h = client.get("/health").json()
        assert h["status"] == "ok" and h["free_slots"] == 2
        r = client.post("/generate", json={"prompt_ids": prompt.tolist(),
                                           "max_new_tokens": 4})
        assert r.status_code == 200
        body = r.json()
        assert body["tokens"] == want


def test_release_idempotent():
    model = make_model()
    eng = ContinuousBatchingEngine(model, max_slots=1, max_seq=64)
    rid = eng.add_request(torch.randint(0, 512, (3,)), 2)
    eng.release(rid)
    eng.release(rid)  # no-op
    assert eng.n_active == 0 and len(eng.free) == 1


def test_per_request_sampling_deterministic():
    """Per-request seeded sampling: same seed -> same tokens, regardless
    of what else shares the batch."""
    model = make_model()
    torch.manual_seed(11)
    prompt = torch.randint(0, 512, (5,))
    eng1 = ContinuousBatchingEngine(model, max_slots=2, max_seq=64)
    r = eng1.add_request(prompt, 4, temperature=0.8, seed=7)
    while eng1.n_active:
        eng1.step()
    alone = eng1.outputs[r]
    eng2 = ContinuousBatchingEngine(model, max_slots=2, max_seq=64)
    r1 = eng2.add_request(prompt, 4, temperature=0.8, seed=7)
    r2 = eng2.add_request(torch.randint(0, 512, (9,)), 6)  # greedy neighbor
    while eng2.n_active:
        eng2.step()
    assert eng2.outputs[r1] == alone


def test_http_sampling_params():
    import threading
    from fastapi.testclient import TestClient
    from hetu_galvatron_amd.cli.serve import build_app
    model = make_model()
    eng = ContinuousBatchingEngine(model, max_slots=1, max_seq=64)
    app = build_app(eng, threading.Lock())
    with TestClient(app) as client:
        b1 = client.post("/generate", json={
            "prompt_ids": [3, 4, 5], "max_new_tokens": 4,
            "temperature": 0.9, "seed": 11}).json()
        b2 = client.post("/generate", json={
            "prompt_ids": [3, 4, 5], "max_new_tokens": 4,
            "temperature": 0.9, "seed": 11}).json()
        assert b1["tokens"] == b2["tokens"]


def test_eos_early_release():
    model = make_model()
    eng = ContinuousBatchingEngine(model, max_slots=1, max_seq=64)
    torch.manual_seed(13)
    prompt = torch.randint(0, 512, (5,))
    rid0 = eng.add_request(prompt, 8)
    while eng.n_active:
        eng.step()
    toks = eng.outputs[rid0]
    # re-run with eos set to the 3rd generated token: generation must
    # stop at the FIRST occurrence of that token
    eos = toks[2]
    rid = eng.add_request(prompt, 8, eos_id=eos)
    steps = 0
    while eng.n_active:
        eng.step()
        steps += 1
        assert steps < 10
    assert eng.outputs[rid] == toks[:toks.index(eos) + 1]


def test_engine_matches_generator_qwen3_and_windowed():
    """The engine's hand-rolled prefill/step must track GalvatronGenerator
    for qk-layernorm models and sliding-window models (the three serving
    gaps: step-side qk norm, window in prefill/decode, interleaved rows)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.inference import GalvatronGenerator
    from hetu_galvatron_amd.runtime.serving import ContinuousBatchingEngine

    for name, extra in (("tiny-qwen3", {}),
                        ("tiny-llama", {"sliding_window": 8}),
                        ("tiny-llama", {"rotary_interleaved": True})):
        cfg = load_config(base={
            "model": dict({"model_name": name}, **extra),
            "parallel": {"mixed_precision": "fp32"},
            "train": {"global_train_batch_size": 2, "train_iters": 1,
                      "distributed_backend": "gloo"}})
        torch.manual_seed(0)
        model = GalvatronModel(cfg)
        gen = GalvatronGenerator(model, max_batch=1, max_seq=32)
        torch.manual_seed(9)
        ids = torch.randint(0, cfg.model.vocab_size, (12,))
        want = gen.generate(ids.unsqueeze(0), max_new_tokens=4,
                            temperature=0.0)[0, 12:].tolist()
        eng = ContinuousBatchingEngine(model, max_slots=2, max_seq=32)
        rid = eng.add_request(ids, max_new_tokens=4)
        while eng.n_active:
            eng.step()
        assert eng.collect(rid) == want, (name, extra)
