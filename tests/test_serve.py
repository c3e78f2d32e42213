"""Inference-service example: /generate must round-trip through the
KV-cached decoder and agree with a direct generate() call."""
import os
import sys

import torch
from fastapi.testclient import TestClient

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "examples"))


def test_serve_generate_roundtrip():
    torch.manual_seed(0)
    from serve_llama import build_app

    app = build_app(config="tiny")
    # context manager: shuts the client's portal threads down so later
    # fork-based tests don't inherit live-thread state
    with TestClient(app) as client:
        h = client.get("/health").json()
        assert h["status"] == "ok" and h["params"] > 0

        r = client.post("/generate", json={"tokens": [5, 6, 7],
                                           "max_new_tokens": 4})
        assert r.status_code == 200
        body = r.json()
        assert body["tokens"][:3] == [5, 6, 7]
        assert len(body["new_tokens"]) == 4

        # greedy determinism: same prompt -> same continuation
        r2 = client.post("/generate", json={"tokens": [5, 6, 7],
                                            "max_new_tokens": 4})
        assert r2.json()["new_tokens"] == body["new_tokens"]

        # validation: empty prompt rejected
        bad = client.post("/generate", json={"tokens": [],
                                             "max_new_tokens": 4})
        assert bad.status_code == 422
