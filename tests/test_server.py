"""Serving-layer test (CPU: NumPy backend, tiny preset)."""

import pytest


def test_completions_endpoint():
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from llm_np_cp_amd.runtime.server import build_app

    app = build_app("tiny-llama", backend="numpy")
    client = TestClient(app)

    r = client.get("/health")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"

    r = client.post("/v1/completions", json={
        "prompt": "Once upon a time", "max_tokens": 4,
        "strategy": "greedy", "stop_on_eos": False})
    assert r.status_code == 200
    body = r.json()
    assert body["usage"]["completion_tokens"] == 4
    assert isinstance(body["choices"][0]["text"], str)
    assert body["timings"]["decode_tokens_per_s"] > 0
