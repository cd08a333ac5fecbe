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


def test_real_socket_two_node_network(tmp_path):
    """Two nodes over REAL HTTP sockets: connect_to_network (seed join,
    peer adoption, chain pull) then consensus through the wire."""
    import socket
    import threading
    import time

    def free_port():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    ports = [free_port(), free_port()]
    nodes = []
    for i, port in enumerate(ports):
        chain = MemoryChain(node_id=f"sock{i}", path=str(tmp_path / f"s{i}.json"),
                            difficulty=1,
                            wallet=FeiCoinWallet(path=str(tmp_path / f"sw{i}.json")))
        node = MemorychainNode(node_id=f"sock{i}", port=port, chain=chain)
        nodes.append(node)
        threading.Thread(target=node.run, daemon=True).start()
    time.sleep(0.8)

    # seed (node 0) already has a block; node 1 joins and adopts it
    nodes[0].chain.add_memory("pre-existing", {"Subject": "before join"})
    # patch peer addresses to 127.0.0.1 (connect_to_network uses localhost)
    assert nodes[1].connect_to_network(f"127.0.0.1:{ports[0]}")
    assert len(nodes[1].chain.blocks) == 2
    assert f"127.0.0.1:{ports[0]}" in nodes[1].chain.nodes

    # node 0 must know node 1 back (register round-trip)
    peer_of_0 = [p for p in nodes[0].chain.nodes if str(ports[1]) in p]
    assert peer_of_0, nodes[0].chain.nodes

    # consensus over real HTTP from node 1
    out = nodes[1].chain.propose_memory("net-mem", {"Subject": "over sockets"})
    assert out["accepted"], out
    assert out["votes"] == 2
    deadline = time.time() + 5
    while time.time() < deadline and len(nodes[0].chain.blocks) < 3:
        time.sleep(0.1)
    assert len(nodes[0].chain.blocks) == 3        # update propagated
    assert nodes[0].chain.blocks[2].memory_id == "net-mem"
    assert nodes[0].chain.validate_chain()


def test_wallet_transactions_route(node):
    c = node.app.test_client()
    node.chain.wallet.credit(node.node_id, 5.0, reason="test reward")
    r = c.get("/memorychain/wallet/transactions")
    assert r.status_code == 200
    txs = r.get_json()["transactions"]
    assert any(t.get("reason") == "test reward" for t in txs)
    r = c.get(f"/memorychain/wallet/transactions?node={node.node_id}&limit=1")
    assert len(r.get_json()["transactions"]) == 1
