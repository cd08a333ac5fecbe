"""memdir HTTP server tests via the Flask test client (route parity:
SURVEY.md §2.3 / server.py)."""

import pytest

from fei_amd.memdir import utils as mu
from fei_amd.memdir.server import create_app


@pytest.fixture
def client(memdir_base):
    app = create_app(base=memdir_base, api_key="secret")
    app.testing = True
    return app.test_client()


def _h(key="secret"):
    return {"X-API-Key": key}


def test_health_no_auth(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert r.get_json()["status"] == "ok"


def test_auth_required(client):
    assert client.get("/memories").status_code == 401
    assert client.get("/memories", headers=_h("wrong")).status_code == 401
    assert client.get("/memories", headers=_h()).status_code == 200


def test_create_and_get_memory(client):
    r = client.post("/memories", json={
        "headers": {"Subject": "via http", "Tags": "api"},
        "body": "posted body", "flags": "F",
    }, headers=_h())
    assert r.status_code == 201
    filename = r.get_json()["filename"]
    unique = mu.parse_memory_filename(filename)["unique"]

    r = client.get(f"/memories/{unique}", headers=_h())
    assert r.status_code == 200
    mem = r.get_json()
    assert mem["headers"]["Subject"] == "via http"
    assert mem["content"] == "posted body"


def test_move_reflag_delete(client):
    r = client.post("/memories", json={"headers": {"Subject": "m"}, "body": ""},
                    headers=_h())
    unique = mu.parse_memory_filename(r.get_json()["filename"])["unique"]

    r = client.put(f"/memories/{unique}", json={"folder": ".Projects"}, headers=_h())
    assert r.get_json()["success"]

    r = client.put(f"/memories/{unique}", json={"flags": "SP"}, headers=_h())
    assert r.get_json()["success"]
    assert r.get_json()["filename"].endswith(":2,PS")

    r = client.delete(f"/memories/{unique}", headers=_h())
    assert r.get_json()["folder"] == ".Trash"

    r = client.delete(f"/memories/{unique}?permanent=1", headers=_h())
    assert r.get_json()["permanent"] is True
    assert client.get(f"/memories/{unique}", headers=_h()).status_code == 404


def test_search_route(client):
    client.post("/memories", json={"headers": {"Subject": "findable", "Tags": "x"},
                                   "body": "alpha beta"}, headers=_h())
    r = client.get("/search?q=%23x+with_content", headers=_h())
    data = r.get_json()
    assert data["count"] == 1
    assert data["results"][0]["content"] == "alpha beta"


def test_folder_routes(client):
    r = client.post("/folders", json={"name": ".Custom"}, headers=_h())
    assert r.status_code == 201
    r = client.get("/folders", headers=_h())
    assert any(f["folder"] == ".Custom" for f in r.get_json()["folders"])
    r = client.get("/folders/.Custom/stats", headers=_h())
    assert r.get_json()["total"] == 0
    r = client.delete("/folders/.Custom", headers=_h())
    assert r.get_json()["success"]


def test_filters_run_route(client):
    client.post("/memories", json={"headers": {"Subject": "todo: thing"},
                                   "body": ""}, headers=_h())
    r = client.post("/filters/run", headers=_h())
    assert r.status_code == 200
    assert r.get_json()["processed"] >= 1
