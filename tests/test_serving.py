"""Serving endpoint (examples/serve_gpt2.py) over the FastAPI test
client: health, id-based generation, validation errors. Uses a tiny
GPT-2-shaped model so it runs in CPU CI."""

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from tnn_amd.nn import LayerBuilder
from tnn_amd import models


@pytest.fixture(scope="module")
def client(tmp_path_factory):
    import examples.serve_gpt2 as srv
    # shrink: register a tiny gpt2 so the test is fast
    from tnn_amd.models import register_model

    @register_model("tiny_gpt2_serving")
    def tiny(dtype=torch.float32):
        return (LayerBuilder((16,), dtype)
                .embedding(128, 32)
                .positional_embedding(64, 32)
                .gpt_block(32, 2, flash=False)
                .layernorm()
                .dense(128, True, "head")
                .build("tiny_gpt2_serving"))

    app = srv.build_app("tiny_gpt2_serving", seq_len=64, device="cpu")
    return TestClient(app)


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"


def test_generate_ids(client):
    r = client.post("/generate", json={"ids": [1, 2, 3],
                                       "max_new_tokens": 4})
    assert r.status_code == 200
    body = r.json()
    assert body["ids"][:3] == [1, 2, 3]
    assert len(body["new_ids"]) >= 1
    assert all(0 <= t < 128 for t in body["new_ids"])
    # deterministic greedy: same request -> same output
    r2 = client.post("/generate", json={"ids": [1, 2, 3],
                                        "max_new_tokens": 4})
    assert r2.json()["ids"] == body["ids"]


def test_generate_validation(client):
    assert client.post("/generate", json={}).status_code == 400
    assert client.post("/generate",
                       json={"text": "hi"}).status_code == 400  # no vocab
    assert client.post("/generate", json={"ids": []}).status_code == 400
