"""HTTP serving API tests (fastapi TestClient; CPU tiny engine)."""

import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from fei_amd.serve.api import create_app


@pytest.fixture(scope="module")
def client():
    app = create_app(model="llama3-tiny")
    return TestClient(app)


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert r.json()["model"] == "llama3-tiny"


def test_models(client):
    r = client.get("/v1/models")
    ids = [m["id"] for m in r.json()["data"]]
    assert "llama3-8b" in ids and "llama3-tiny" in ids


def test_completions_greedy(client):
    r = client.post("/v1/completions", json={
        "prompt": "abc abc abc", "max_tokens": 8})
    assert r.status_code == 200
    body = r.json()
    assert body["choices"][0]["text"]
    assert body["usage"]["completion_tokens"] >= 1
    assert "decode_tok_s" in body["metrics"]


def test_completions_sampled(client):
    r = client.post("/v1/completions", json={
        "prompt": "xyz", "max_tokens": 8, "temperature": 0.9, "top_k": 20})
    assert r.status_code == 200
    assert r.json()["choices"][0]["text"] is not None


def test_chat_completions(client):
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "system", "content": "be brief"},
                     {"role": "user", "content": "hello"}],
        "max_tokens": 8})
    assert r.status_code == 200
    msg = r.json()["choices"][0]["message"]
    assert msg["role"] == "assistant"


def test_validation_error(client):
    r = client.post("/v1/completions", json={"max_tokens": 4})
    assert r.status_code == 422          # prompt is required
