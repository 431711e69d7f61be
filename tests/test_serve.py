"""Inference server: /generate over the KV-cached path (TestClient, no
network)."""

import torch


def test_serve_generate_endpoint():
    from fastapi.testclient import TestClient

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.serve import build_app

    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=128, n_positions=64, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    stage.eval()
    client = TestClient(build_app(stage))

    assert client.get("/health").json()["status"] == "ok"
    r = client.post("/generate", json={"input_ids": [1, 2, 3],
                                       "max_new_tokens": 5})
    assert r.status_code == 200
    body = r.json()
    assert len(body["new_ids"]) == 5
    assert body["output_ids"][:3] == [1, 2, 3]
    # greedy must equal the direct call
    direct = stage.generate(torch.tensor([[1, 2, 3]]), max_new_tokens=5)
    assert body["output_ids"] == direct[0].tolist()
    # sampling + int8 cache path
    r2 = client.post("/generate", json={"input_ids": [4, 5], "max_new_tokens": 4,
                                        "temperature": 0.8, "top_k": 10,
                                        "cache_dtype": "int8"})
    assert r2.status_code == 200 and len(r2.json()["new_ids"]) == 4


def test_serve_speculative_endpoint():
    import torch
    from fastapi.testclient import TestClient

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.serve import build_app

    torch.manual_seed(0)
    base = dict(vocab_size=96, n_positions=64, dropout=0.0)
    target = GPT2Stage(GPT2Config(n_embd=64, n_layer=2, n_head=2, **base)).eval()
    draft = GPT2Stage(GPT2Config(n_embd=32, n_layer=1, n_head=2, **base)).eval()
    client = TestClient(build_app(target, draft=draft))
    ids = [3, 5, 7, 9]
    r = client.post("/generate", json={
        "input_ids": ids, "max_new_tokens": 8, "speculative": True,
    })
    assert r.status_code == 200
    spec = r.json()["new_ids"]
    plain = client.post("/generate", json={
        "input_ids": ids, "max_new_tokens": 8,
    }).json()["new_ids"]
    assert spec == plain  # greedy speculative is exact


def test_serve_beam_endpoint():
    import torch
    from fastapi.testclient import TestClient

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.serve import build_app

    torch.manual_seed(1)
    stage = GPT2Stage(GPT2Config(n_embd=64, n_layer=2, n_head=2,
                                 vocab_size=96, n_positions=64,
                                 dropout=0.0)).eval()
    client = TestClient(build_app(stage))
    r = client.post("/generate", json={
        "input_ids": [1, 2, 3], "max_new_tokens": 6, "num_beams": 3,
    })
    assert r.status_code == 200
    assert len(r.json()["new_ids"]) == 6
