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


def test_sessions_endpoints(client):
    a = client.post("/v1/sessions", json={"prompt": "abc abc", "max_tokens": 6}).json()
    b = client.post("/v1/sessions", json={"prompt": "xyz xyz", "max_tokens": 6}).json()
    sid_a, sid_b = a["session_id"], b["session_id"]
    for _ in range(8):
        st = client.post("/v1/sessions/step?n=2").json()
        if st["active"] == 0:
            break
    ra = client.get(f"/v1/sessions/{sid_a}").json()
    rb = client.get(f"/v1/sessions/{sid_b}").json()
    assert ra["done"] and rb["done"]
    assert len(ra["token_ids"]) >= 1 and len(rb["token_ids"]) >= 1
    assert client.delete(f"/v1/sessions/{sid_a}").json()["closed"] == sid_a
    client.delete(f"/v1/sessions/{sid_b}")
    assert client.get(f"/v1/sessions/{sid_a}").status_code == 404


def test_completions_stream_sse(client):
    with client.stream("POST", "/v1/completions", json={
            "prompt": "stream abc", "max_tokens": 12, "stream": True,
            "stop_on_eos": False}) as r:
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/event-stream")
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    import json
    chunks = [json.loads(l[6:]) for l in lines[:-1]]
    assert chunks and chunks[-1]["choices"][0]["finish_reason"] == "stop"
    text = "".join(c["choices"][0]["text"] for c in chunks)
    assert len(text) >= 1


def test_chat_stream_sse(client):
    with client.stream("POST", "/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 10, "stream": True}) as r:
        assert r.status_code == 200
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"


def test_sessions_503_on_pool_exhaustion():
    from fei_amd.engine.engine import LocalEngine
    from fei_amd.serve.api import create_app
    eng = LocalEngine.create("llama3-tiny")
    c = TestClient(create_app(engine=eng, session_blocks=3))
    r1 = c.post("/v1/sessions", json={"prompt": "abcdefgh" * 3,
                                      "max_tokens": 2})
    assert r1.status_code == 200
    r2 = c.post("/v1/sessions", json={"prompt": "ijklmnop" * 5,
                                      "max_tokens": 2})
    assert r2.status_code == 503          # admission control over HTTP
    c.delete(f"/v1/sessions/{r1.json()['session_id']}")
    r3 = c.post("/v1/sessions", json={"prompt": "ijklmnop" * 5,
                                      "max_tokens": 2})
    assert r3.status_code == 200          # eviction freed the pool


def test_completions_eval_mode_logprobs(client):
    r = client.post("/v1/completions", json={
        "prompt": "evaluate this text", "max_tokens": 0, "echo": True,
        "logprobs": 0})
    assert r.status_code == 200
    lp = r.json()["choices"][0]["logprobs"]
    assert lp["token_logprobs"][0] is None
    assert all(isinstance(x, float) and x <= 0 for x in lp["token_logprobs"][1:])


def test_concurrent_completions_serialized(client):
    """Parallel HTTP requests must serialize cleanly on the engine lock
    (no interleaved decode state, every response well-formed)."""
    import concurrent.futures as cf

    def hit(i):
        r = client.post("/v1/completions", json={
            "prompt": f"concurrent request {i}", "max_tokens": 6})
        assert r.status_code == 200
        return r.json()["choices"][0]["text"]

    with cf.ThreadPoolExecutor(max_workers=4) as ex:
        texts = list(ex.map(hit, range(8)))
    assert len(texts) == 8
    assert all(isinstance(t, str) for t in texts)


def test_cross_request_prefix_cache(client):
    p1 = "shared long preamble " * 20 + "question one"
    client.post("/v1/completions", json={"prompt": p1, "max_tokens": 4})
    r = client.post("/v1/completions", json={
        "prompt": p1 + " and question two", "max_tokens": 4})
    eng = client.app.state.engine
    assert eng.last_metrics["cached_prefix"] > 50     # reused the preamble
    # opt-out works
    r = client.post("/v1/completions", json={
        "prompt": p1, "max_tokens": 4, "cache_prefix": False})
    assert r.status_code == 200
    assert eng.last_metrics["cached_prefix"] == 0


def test_stop_and_n(client):
    r = client.post("/v1/completions", json={
        "prompt": "abc abc abc", "max_tokens": 16, "n": 2,
        "stop_on_eos": False})
    body = r.json()
    assert len(body["choices"]) == 2
    assert body["choices"][1]["index"] == 1
    r = client.post("/v1/completions", json={
        "prompt": "abc abc abc", "max_tokens": 16, "stop": ["zzqq"],
        "stop_on_eos": False})
    assert r.status_code == 200


def test_api_main_check_flag(capsys):
    from fei_amd.serve.api import main
    assert main(["--model", "llama3-tiny", "--check"]) == 0
    assert "ok: llama3-tiny" in capsys.readouterr().out
