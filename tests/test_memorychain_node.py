"""Memorychain HTTP node tests via Flask test clients, including a 2-node
consensus over test-client transports (no sockets)."""

import pytest

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.memorychain.node import MemorychainNode
from fei_amd.memorychain.wallet import FeiCoinWallet


def make_node(tmp_path, name, port):
    chain = MemoryChain(node_id=name, path=str(tmp_path / f"{name}.json"),
                        difficulty=1,
                        wallet=FeiCoinWallet(path=str(tmp_path / f"{name}_w.json")))
    node = MemorychainNode(node_id=name, port=port, chain=chain)
    node.app.testing = True
    return node


@pytest.fixture
def node(tmp_path):
    return make_node(tmp_path, "n0", 6789)


def test_health_and_status(node):
    c = node.app.test_client()
    assert c.get("/memorychain/health").get_json()["status"] == "ok"
    st = c.get("/memorychain/node_status").get_json()
    assert st["node_id"] == "n0"
    assert st["chain_length"] == 1
    assert "ai_model" in st


def test_propose_and_chain(node):
    c = node.app.test_client()
    r = c.post("/memorychain/propose",
               json={"memory_id": "m1", "memory_data": {"Subject": "via http"}})
    assert r.status_code == 200 and r.get_json()["accepted"]
    chain = c.get("/memorychain/chain").get_json()
    assert chain["length"] == 2
    r = c.post("/memorychain/propose", json={"memory_id": "m1",
                                             "memory_data": {"Subject": "dup"}})
    assert r.status_code == 422


def test_task_routes(node):
    c = node.app.test_client()
    r = c.post("/memorychain/propose_task",
               json={"task_id": "t1", "task_data": {"Subject": "do it"},
                     "reward": 2.5})
    assert r.get_json()["accepted"]
    assert c.post("/memorychain/claim_task", json={"task_id": "t1"}).get_json()["ok"]
    assert c.post("/memorychain/submit_solution",
                  json={"task_id": "t1", "solution": "done"}).get_json()["ok"]
    r = c.post("/memorychain/vote_solution",
               json={"task_id": "t1", "solution_index": 0, "approve": True})
    assert r.get_json()["completed"]
    r = c.get("/memorychain/wallet/balance?node=n0")
    assert r.get_json()["balance"] == 2.5
    tasks = c.get("/memorychain/tasks?state=completed").get_json()["tasks"]
    assert len(tasks) == 1
    r = c.get("/memorychain/tasks/t1")
    assert r.get_json()["memory_id"] == "t1"
    assert c.get("/memorychain/tasks/nope").status_code == 404


def test_register_and_sync(node):
    c = node.app.test_client()
    r = c.post("/memorychain/register", json={"address": "host1:7000"})
    assert r.get_json()["ok"]
    r = c.post("/memorychain/sync_nodes", json={"nodes": ["host2:7000"]})
    assert set(r.get_json()["nodes"]) == {"host1:7000", "host2:7000"}


def test_two_node_http_consensus(tmp_path):
    """Wire two nodes' transports through Flask test clients — consensus and
    chain sync over the real routes with no sockets."""
    n0 = make_node(tmp_path, "n0", 6789)
    n1 = make_node(tmp_path, "n1", 6790)
    clients = {"addr0": n0.app.test_client(), "addr1": n1.app.test_client()}

    def vote(peer, proposal):
        r = clients[peer].post("/memorychain/vote", json=proposal)
        return bool(r.get_json().get("vote"))

    def update(peer, chain):
        r = clients[peer].post("/memorychain/update", json={"chain": chain})
        return bool(r.get_json().get("accepted"))

    n0.chain.vote_transport = vote
    n0.chain.update_transport = update
    n0.chain.register_node("addr1")

    r = clients["addr0"].post("/memorychain/propose",
                              json={"memory_id": "mnet",
                                    "memory_data": {"Subject": "federated"}})
    out = r.get_json()
    assert out["accepted"] and out["votes"] == 2
    # n1 received the chain update through its /update route
    assert len(n1.chain.blocks) == 2
    assert n1.chain.blocks[1].memory_id == "mnet"


def test_responsible_memories_route(node):
    c = node.app.test_client()
    c.post("/memorychain/propose",
           json={"memory_id": "m1", "memory_data": {"Subject": "a"}})
    r = c.get("/memorychain/responsible_memories")
    data = r.get_json()
    assert data["node"] == "n0"
    assert len(data["memories"]) == 1   # single node owns everything


def test_update_status_route(node):
    c = node.app.test_client()
    c.post("/memorychain/update_status",
           json={"status": "working", "load": 0.7, "current_task": "t9"})
    st = c.get("/memorychain/node_status").get_json()
    assert st["status"] == "working" and st["load"] == 0.7
    assert st["current_task"] == "t9"
