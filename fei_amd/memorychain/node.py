"""Memorychain node: the HTTP control plane.

Route parity with the reference Flask app (memdir_tools/memorychain.py:
1263-1685) under ``/memorychain/*``:
  vote, update, propose, propose_task, claim_task, submit_solution,
  vote_solution, vote_difficulty, wallet/balance, wallet/transfer,
  wallet/transactions,
  register, sync_nodes, chain, tasks, tasks/<id>, network_status,
  responsible_memories, health, node_status, update_status

Node status carries ai_model/load/current_task fields
(reference: memorychain.py:1249-1255). ``connect_to_network`` joins via a
seed node (reference: memorychain.py:1726-1765).
"""

from __future__ import annotations

import threading
import time
from typing import Any, Dict, List, Optional

import requests
from flask import Flask, jsonify, request

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.utils.logging import get_logger

logger = get_logger("memorychain.node")

VOTE_TIMEOUT_S = 5
UPDATE_TIMEOUT_S = 10


def http_vote_transport(peer: str, proposal: Dict[str, Any]) -> bool:
    """POST /memorychain/vote; failures count as a 'no' vote
    (reference: memorychain.py:988-1001)."""
    try:
        r = requests.post(f"http://{peer}/memorychain/vote", json=proposal,
                          timeout=VOTE_TIMEOUT_S)
        return bool(r.ok and r.json().get("vote"))
    except requests.RequestException:
        return False


def http_update_transport(peer: str, chain: List[Dict[str, Any]]) -> bool:
    try:
        r = requests.post(f"http://{peer}/memorychain/update",
                          json={"chain": chain}, timeout=UPDATE_TIMEOUT_S)
        return bool(r.ok and r.json().get("accepted"))
    except requests.RequestException:
        return False


class MemorychainNode:
    def __init__(
        self,
        node_id: str,
        port: int = 6789,
        chain: Optional[MemoryChain] = None,
        chain_path: Optional[str] = None,
        difficulty: int = 2,
        ai_model: str = "llama3-8b-local",
        persist: bool = True,
    ):
        self.node_id = node_id
        self.port = port
        self.ai_model = ai_model
        self.load = 0.0
        self.current_task: Optional[str] = None
        self.status = "idle"
        self.chain = chain or MemoryChain(
            node_id=node_id, path=chain_path, difficulty=difficulty,
            vote_transport=http_vote_transport,
            update_transport=http_update_transport, persist=persist)
        if chain is not None:
            # a caller-provided chain serves THIS node: give it the HTTP
            # transports (callers may still override afterwards, e.g. the
            # test-client harness or the xGMI collective path)
            self.chain.vote_transport = http_vote_transport
            self.chain.update_transport = http_update_transport
        self.app = self._build_app()
        self._server_thread: Optional[threading.Thread] = None

    # -- network -------------------------------------------------------------

    def connect_to_network(self, seed: str) -> bool:
        """Register with a seed node, adopt its peer list + chain
        (reference: memorychain.py:1726-1765)."""
        me = f"localhost:{self.port}"
        try:
            r = requests.post(f"http://{seed}/memorychain/register",
                              json={"address": me}, timeout=UPDATE_TIMEOUT_S)
            if not r.ok:
                return False
            data = r.json()
            self.chain.register_node(seed)
            for peer in data.get("nodes", []):
                if peer != me:
                    self.chain.register_node(peer)
            rc = requests.get(f"http://{seed}/memorychain/chain",
                              timeout=UPDATE_TIMEOUT_S)
            if rc.ok:
                self.chain.receive_chain_update(rc.json().get("chain", []))
            # announce ourselves to the other peers
            for peer in list(self.chain.nodes):
                if peer != seed:
                    try:
                        requests.post(f"http://{peer}/memorychain/register",
                                      json={"address": me}, timeout=VOTE_TIMEOUT_S)
                    except requests.RequestException:
                        pass
            return True
        except requests.RequestException as e:
            logger.warning("connect_to_network failed: %s", e)
            return False

    def node_status(self) -> Dict[str, Any]:
        return {
            "node_id": self.node_id,
            "address": f"localhost:{self.port}",
            "ai_model": self.ai_model,
            "load": self.load,
            "status": self.status,
            "current_task": self.current_task,
            "chain_length": len(self.chain.blocks),
            "peers": list(self.chain.nodes),
        }

    # -- flask app -----------------------------------------------------------

    def _build_app(self) -> Flask:
        app = Flask(f"memorychain-{self.node_id}")
        chain = self.chain
        node = self

        @app.get("/memorychain/health")
        def health():
            return jsonify({"status": "ok", "node_id": node.node_id})

        @app.post("/memorychain/vote")
        def vote():
            proposal = request.get_json(force=True, silent=True) or {}
            return jsonify({"vote": chain.vote_on_proposal(proposal)})

        @app.post("/memorychain/update")
        def update():
            data = request.get_json(force=True, silent=True) or {}
            return jsonify({"accepted": chain.receive_chain_update(
                data.get("chain", []))})

        @app.post("/memorychain/propose")
        def propose():
            data = request.get_json(force=True, silent=True) or {}
            result = chain.propose_memory(
                data.get("memory_id") or f"mem-{int(time.time()*1000)}",
                data.get("memory_data", data.get("memory", {})))
            code = 200 if result.get("accepted") else 422
            return jsonify(result), code

        @app.post("/memorychain/propose_task")
        def propose_task():
            data = request.get_json(force=True, silent=True) or {}
            result = chain.propose_task(
                data.get("task_id") or f"task-{int(time.time()*1000)}",
                data.get("task_data", {}),
                reward=float(data.get("reward", 1.0)),
                difficulty=int(data.get("difficulty", 1)))
            return jsonify(result), 200 if result.get("accepted") else 422

        @app.post("/memorychain/claim_task")
        def claim_task():
            data = request.get_json(force=True, silent=True) or {}
            ok = chain.claim_task(data.get("task_id", ""),
                                  data.get("node_id"))
            if ok:
                node.current_task = data.get("task_id")
                node.status = "working"
            return jsonify({"ok": ok})

        @app.post("/memorychain/submit_solution")
        def submit_solution():
            data = request.get_json(force=True, silent=True) or {}
            ok = chain.submit_solution(data.get("task_id", ""),
                                       data.get("solution", ""),
                                       data.get("node_id"))
            return jsonify({"ok": ok})

        @app.post("/memorychain/vote_solution")
        def vote_solution():
            data = request.get_json(force=True, silent=True) or {}
            return jsonify(chain.vote_on_solution(
                data.get("task_id", ""), int(data.get("solution_index", 0)),
                bool(data.get("approve", True)), data.get("voter")))

        @app.post("/memorychain/vote_difficulty")
        def vote_difficulty():
            data = request.get_json(force=True, silent=True) or {}
            return jsonify(chain.vote_on_task_difficulty(
                data.get("task_id", ""), int(data.get("difficulty", 1)),
                data.get("voter")))

        @app.get("/memorychain/wallet/balance")
        def wallet_balance():
            who = request.args.get("node", node.node_id)
            return jsonify({"node": who, "balance": chain.wallet.balance(who)})

        @app.get("/memorychain/wallet/transactions")
        def wallet_transactions():
            who = request.args.get("node")
            txs = chain.wallet.transactions
            if who:
                txs = [t for t in txs
                       if t.get("from") == who or t.get("to") == who or
                       t.get("node") == who]
            limit = int(request.args.get("limit", 50))
            return jsonify({"transactions": txs[-limit:]})

        @app.post("/memorychain/wallet/transfer")
        def wallet_transfer():
            data = request.get_json(force=True, silent=True) or {}
            ok = chain.wallet.transfer(data.get("from", ""), data.get("to", ""),
                                       float(data.get("amount", 0)))
            return jsonify({"ok": ok})

        @app.post("/memorychain/register")
        def register():
            data = request.get_json(force=True, silent=True) or {}
            addr = data.get("address", "")
            known = list(chain.nodes)
            chain.register_node(addr)
            return jsonify({"ok": True, "nodes": known})

        @app.post("/memorychain/sync_nodes")
        def sync_nodes():
            data = request.get_json(force=True, silent=True) or {}
            for addr in data.get("nodes", []):
                chain.register_node(addr)
            return jsonify({"ok": True, "nodes": list(chain.nodes)})

        @app.get("/memorychain/chain")
        def get_chain():
            return jsonify({"chain": chain.serialize(),
                            "length": len(chain.blocks)})

        @app.get("/memorychain/tasks")
        def tasks():
            state = request.args.get("state")
            return jsonify({"tasks": chain.tasks(state)})

        @app.get("/memorychain/tasks/<task_id>")
        def task(task_id):
            for t in chain.tasks():
                if t["memory_id"] == task_id:
                    return jsonify(t)
            return jsonify({"error": "not found"}), 404

        @app.get("/memorychain/network_status")
        def network_status():
            peers = {}
            for peer in list(chain.nodes):
                try:
                    r = requests.get(f"http://{peer}/memorychain/node_status",
                                     timeout=VOTE_TIMEOUT_S)
                    peers[peer] = r.json() if r.ok else {"status": "error"}
                except requests.RequestException:
                    peers[peer] = {"status": "unreachable"}
            return jsonify({"self": node.node_status(), "peers": peers,
                            "chain": chain.stats()})

        @app.get("/memorychain/responsible_memories")
        def responsible_memories():
            who = request.args.get("node", node.node_id)
            mine = [b.to_dict() for b in chain.blocks[1:]
                    if chain.responsible_node(b.memory_id) == who]
            return jsonify({"node": who, "memories": mine})

        @app.get("/memorychain/node_status")
        def node_status():
            return jsonify(node.node_status())

        @app.post("/memorychain/update_status")
        def update_status():
            data = request.get_json(force=True, silent=True) or {}
            node.status = data.get("status", node.status)
            node.load = float(data.get("load", node.load))
            node.current_task = data.get("current_task", node.current_task)
            return jsonify({"ok": True})

        return app

    # -- lifecycle -----------------------------------------------------------

    def run(self, host: str = "127.0.0.1", threaded: bool = True) -> None:
        self.app.run(host=host, port=self.port, threaded=threaded)

    def start_background(self, host: str = "127.0.0.1") -> None:
        t = threading.Thread(target=self.run, kwargs={"host": host}, daemon=True)
        t.start()
        self._server_thread = t
