"""Memorychain core: blocks, chain, consensus and task ledger logic.

Parity with the reference (memdir_tools/memorychain.py):
  - MemoryBlock: SHA-256 block hash over index/timestamp/memory-id/prev/
    proposer/task fields (:110-130), toy leading-zeros proof-of-work
    (:132-143), task state machine proposed -> accepted -> in_progress ->
    solution_proposed -> completed/rejected (:58-63,145-261), JSON
    serialization (:263-327)
  - MemoryChain: genesis block (:528-550), append+mine add_memory
    (:562-594), hash+link validate_chain (:596-618), 51 %-quorum
    propose_memory with a thread-pool vote broadcast (:620-685;
    MIN_QUORUM_PERCENT :54), responsible node = hash(id) mod nodes
    (:668-671), task ops incl. reward payout (:687-913), proposal
    validation rules (:932-965), longest-chain-wins sync requiring the own
    chain to be a prefix (:1037-1085), JSON persistence (:1140-1172)

Redesigned for testability: the vote/update transports are injected
callables (HTTP in node.py, direct calls in the in-process test harness —
the integration-test gap SURVEY.md §4 flags in the reference).
"""

from __future__ import annotations

import hashlib
import json
import os
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Callable, Dict, List, Optional

from fei_amd.memorychain.wallet import FeiCoinWallet
from fei_amd.utils.logging import get_logger

logger = get_logger("memorychain.chain")

MIN_QUORUM_PERCENT = 51
DEFAULT_DIFFICULTY = 2
DEFAULT_CHAIN_PATH = os.path.join(os.path.expanduser("~"), ".memdir",
                                  "memorychain.json")


class TaskState:
    PROPOSED = "proposed"
    ACCEPTED = "accepted"
    IN_PROGRESS = "in_progress"
    SOLUTION_PROPOSED = "solution_proposed"
    COMPLETED = "completed"
    REJECTED = "rejected"

    ORDER = [PROPOSED, ACCEPTED, IN_PROGRESS, SOLUTION_PROPOSED, COMPLETED]


class MemoryBlock:
    def __init__(
        self,
        index: int,
        timestamp: float,
        memory_id: str,
        memory_data: Dict[str, Any],
        proposer_node: str,
        prev_hash: str,
        nonce: int = 0,
        is_task: bool = False,
        task_state: str = TaskState.PROPOSED,
        working_nodes: Optional[List[str]] = None,
        solutions: Optional[List[Dict[str, Any]]] = None,
        difficulty: int = 1,
        difficulty_votes: Optional[Dict[str, int]] = None,
        reward: float = 0.0,
        block_hash: Optional[str] = None,
    ):
        self.index = index
        self.timestamp = timestamp
        self.memory_id = memory_id
        self.memory_data = memory_data
        self.proposer_node = proposer_node
        self.prev_hash = prev_hash
        self.nonce = nonce
        self.is_task = is_task
        self.task_state = task_state
        self.working_nodes = working_nodes or []
        self.solutions = solutions or []
        self.difficulty = difficulty
        self.difficulty_votes = difficulty_votes or {}
        self.reward = reward
        self.hash = block_hash or self.compute_hash()

    # -- hashing / mining ----------------------------------------------------

    def compute_hash(self) -> str:
        payload = json.dumps({
            "index": self.index,
            "timestamp": self.timestamp,
            "memory_id": self.memory_id,
            "memory_data": self.memory_data,
            "proposer_node": self.proposer_node,
            "prev_hash": self.prev_hash,
            "nonce": self.nonce,
            "is_task": self.is_task,
            "task_state": self.task_state,
            "working_nodes": self.working_nodes,
            "difficulty": self.difficulty,
        }, sort_keys=True)
        return hashlib.sha256(payload.encode("utf-8")).hexdigest()

    def mine_block(self, difficulty: int = DEFAULT_DIFFICULTY) -> None:
        """Toy PoW: find a nonce whose hash has `difficulty` leading zeros.
        Kept on CPU deliberately (do not GPU-accelerate a
        toy)."""
        target = "0" * difficulty
        while not self.hash.startswith(target):
            self.nonce += 1
            self.hash = self.compute_hash()

    # -- serialization -------------------------------------------------------

    def to_dict(self) -> Dict[str, Any]:
        return {
            "index": self.index,
            "timestamp": self.timestamp,
            "memory_id": self.memory_id,
            "memory_data": self.memory_data,
            "proposer_node": self.proposer_node,
            "prev_hash": self.prev_hash,
            "nonce": self.nonce,
            "hash": self.hash,
            "is_task": self.is_task,
            "task_state": self.task_state,
            "working_nodes": self.working_nodes,
            "solutions": self.solutions,
            "difficulty": self.difficulty,
            "difficulty_votes": self.difficulty_votes,
            "reward": self.reward,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "MemoryBlock":
        return cls(
            index=d["index"], timestamp=d["timestamp"],
            memory_id=d["memory_id"], memory_data=d.get("memory_data", {}),
            proposer_node=d.get("proposer_node", ""),
            prev_hash=d.get("prev_hash", ""), nonce=d.get("nonce", 0),
            is_task=d.get("is_task", False),
            task_state=d.get("task_state", TaskState.PROPOSED),
            working_nodes=d.get("working_nodes"),
            solutions=d.get("solutions"),
            difficulty=d.get("difficulty", 1),
            difficulty_votes=d.get("difficulty_votes"),
            reward=d.get("reward", 0.0),
            block_hash=d.get("hash"),
        )


VoteTransport = Callable[[str, Dict[str, Any]], bool]
UpdateTransport = Callable[[str, List[Dict[str, Any]]], bool]


class MemoryChain:
    def __init__(
        self,
        node_id: str,
        path: Optional[str] = None,
        difficulty: int = DEFAULT_DIFFICULTY,
        wallet: Optional[FeiCoinWallet] = None,
        vote_transport: Optional[VoteTransport] = None,
        update_transport: Optional[UpdateTransport] = None,
        persist: bool = True,
    ):
        self.node_id = node_id
        self.path = path or DEFAULT_CHAIN_PATH
        self.difficulty = difficulty
        self.persist = persist
        self.wallet = wallet or FeiCoinWallet(
            path=os.path.join(os.path.dirname(self.path), "feicoin_wallet.json")
            if persist else os.devnull)
        self._lock = threading.RLock()
        self.nodes: List[str] = []            # peer addresses (host:port)
        self.vote_transport = vote_transport or (lambda peer, prop: False)
        self.update_transport = update_transport or (lambda peer, chain: False)
        self.blocks: List[MemoryBlock] = []
        if persist and os.path.exists(self.path):
            self._load()
        if not self.blocks:
            self._genesis()

    # -- base chain ops ------------------------------------------------------

    def _genesis(self) -> None:
        g = MemoryBlock(index=0, timestamp=0.0, memory_id="genesis",
                        memory_data={"Subject": "genesis"},
                        proposer_node="genesis", prev_hash="0")
        g.mine_block(1)
        self.blocks = [g]
        self._save()

    def last_block(self) -> MemoryBlock:
        return self.blocks[-1]

    def add_memory(self, memory_id: str, memory_data: Dict[str, Any],
                   proposer: Optional[str] = None,
                   timestamp: Optional[float] = None,
                   **task_kwargs) -> MemoryBlock:
        """Append + mine (no consensus — use propose_memory for that).
        A fixed ``timestamp`` makes the block deterministic across ranks
        (collective federation appends)."""
        with self._lock:
            block = MemoryBlock(
                index=len(self.blocks),
                timestamp=time.time() if timestamp is None else timestamp,
                memory_id=memory_id,
                memory_data=memory_data,
                proposer_node=proposer or self.node_id,
                prev_hash=self.last_block().hash,
                **task_kwargs,
            )
            block.mine_block(self.difficulty)
            self.blocks.append(block)
            self._save()
            return block

    def validate_chain(self, blocks: Optional[List[MemoryBlock]] = None) -> bool:
        chain = blocks if blocks is not None else self.blocks
        for i, b in enumerate(chain):
            if b.hash != b.compute_hash():
                return False
            if i > 0:
                if b.prev_hash != chain[i - 1].hash or b.index != i:
                    return False
                if not b.hash.startswith("0" * min(self.difficulty, 1)):
                    return False
        return True

    # -- consensus -----------------------------------------------------------

    def vote_on_proposal(self, proposal: Dict[str, Any]) -> bool:
        """Validation rules a peer applies to an incoming proposal
        (reference: memorychain.py:932-965): schema + dedupe."""
        memory_id = proposal.get("memory_id")
        data = proposal.get("memory_data")
        if not memory_id or not isinstance(data, dict):
            return False
        if not data.get("Subject") and not data.get("content"):
            return False
        with self._lock:
            if any(b.memory_id == memory_id for b in self.blocks):
                return False
        return True

    def propose_memory(self, memory_id: str, memory_data: Dict[str, Any],
                       **task_kwargs) -> Dict[str, Any]:
        """Quorum consensus: broadcast vote requests (10-thread pool), count
        self-vote, require >=51 %; on approval mine+append and broadcast the
        chain (reference: memorychain.py:620-685)."""
        proposal = {
            "memory_id": memory_id,
            "memory_data": memory_data,
            "proposer_node": self.node_id,
            "timestamp": time.time(),
        }
        peers = list(self.nodes)
        if not peers:
            # single-node fast path (reference: memorychain.py:645-649)
            if not self.vote_on_proposal(proposal):
                return {"accepted": False, "reason": "self-validation failed"}
            block = self.add_memory(memory_id, memory_data, **task_kwargs)
            return {"accepted": True, "votes": 1, "total": 1,
                    "block_index": block.index}
        if not self.vote_on_proposal(proposal):
            return {"accepted": False, "reason": "self-validation failed"}
        votes = 1                                    # self-vote
        total = 1 + len(peers)

        def safe_vote(peer: str) -> bool:
            try:
                return bool(self.vote_transport(peer, proposal))
            except Exception:                        # noqa: BLE001
                return False                         # failure = "no" vote
        with ThreadPoolExecutor(max_workers=10) as pool:
            results = list(pool.map(safe_vote, peers))
        votes += sum(1 for r in results if r)
        accepted = votes * 100 >= MIN_QUORUM_PERCENT * total
        out: Dict[str, Any] = {"accepted": accepted, "votes": votes,
                               "total": total}
        if accepted:
            block = self.add_memory(memory_id, memory_data, **task_kwargs)
            out["block_index"] = block.index
            out["responsible_node"] = self.responsible_node(memory_id)
            self.broadcast_chain_update()
        return out

    def responsible_node(self, memory_id: str) -> str:
        """Deterministic owner: hash(id) mod participants
        (reference: memorychain.py:668-671)."""
        participants = sorted([self.node_id] + list(self.nodes))
        h = int(hashlib.sha256(memory_id.encode()).hexdigest(), 16)
        return participants[h % len(participants)]

    def broadcast_chain_update(self) -> int:
        serialized = self.serialize()
        peers = list(self.nodes)
        if not peers:
            return 0
        def safe_update(peer: str) -> bool:
            try:
                return bool(self.update_transport(peer, serialized))
            except Exception:                        # noqa: BLE001
                return False
        with ThreadPoolExecutor(max_workers=10) as pool:
            results = list(pool.map(safe_update, peers))
        return sum(1 for r in results if r)

    def receive_chain_update(self, blocks_data: List[Dict[str, Any]]) -> bool:
        """Longest-chain-wins with the prefix rule: accept only when the
        incoming chain is longer, valid, and our chain is its prefix
        (reference: memorychain.py:1037-1085 — the rule is racy under
        concurrent proposals; divergent chains are rejected, not healed.
        Preserved behavior, covered by the in-process harness test)."""
        incoming = [MemoryBlock.from_dict(d) for d in blocks_data]
        with self._lock:
            if len(incoming) <= len(self.blocks):
                return False
            if not self.validate_chain(incoming):
                return False
            for ours, theirs in zip(self.blocks, incoming):
                if ours.hash != theirs.hash:
                    return False                     # divergence: reject
            self.blocks = incoming
            self._save()
            return True

    # -- task ledger ---------------------------------------------------------

    def propose_task(self, memory_id: str, task_data: Dict[str, Any],
                     reward: float = 1.0, difficulty: int = 1) -> Dict[str, Any]:
        return self.propose_memory(memory_id, task_data, is_task=True,
                                   task_state=TaskState.ACCEPTED,
                                   reward=reward, difficulty=difficulty)

    def _find_task(self, task_id: str) -> Optional[MemoryBlock]:
        for b in reversed(self.blocks):
            if b.is_task and b.memory_id == task_id:
                return b
        return None

    def claim_task(self, task_id: str, node_id: Optional[str] = None) -> bool:
        node_id = node_id or self.node_id
        with self._lock:
            b = self._find_task(task_id)
            if b is None or b.task_state in (TaskState.COMPLETED, TaskState.REJECTED):
                return False
            if node_id not in b.working_nodes:
                b.working_nodes.append(node_id)
            if b.task_state == TaskState.ACCEPTED:
                b.task_state = TaskState.IN_PROGRESS
            b.hash = b.compute_hash()
            b.mine_block(self.difficulty)
            self._rehash_from(b.index + 1)
            self._save()
            return True

    def submit_solution(self, task_id: str, solution: str,
                        node_id: Optional[str] = None) -> bool:
        node_id = node_id or self.node_id
        with self._lock:
            b = self._find_task(task_id)
            if b is None or b.task_state in (TaskState.COMPLETED, TaskState.REJECTED):
                return False
            b.solutions.append({"node": node_id, "solution": solution,
                                "ts": time.time(), "votes": []})
            b.task_state = TaskState.SOLUTION_PROPOSED
            b.hash = b.compute_hash()
            b.mine_block(self.difficulty)
            self._rehash_from(b.index + 1)
            self._save()
            return True

    def vote_on_solution(self, task_id: str, solution_index: int,
                         approve: bool, voter: Optional[str] = None) -> Dict[str, Any]:
        """Majority of known participants approves -> COMPLETED + reward
        payout to the solver (reference: memorychain.py:829-850)."""
        voter = voter or self.node_id
        with self._lock:
            b = self._find_task(task_id)
            if b is None or solution_index >= len(b.solutions):
                return {"ok": False, "reason": "no such task/solution"}
            sol = b.solutions[solution_index]
            votes = sol.setdefault("votes", [])
            votes[:] = [v for v in votes if v["voter"] != voter]
            votes.append({"voter": voter, "approve": approve})
            participants = max(1 + len(self.nodes), 1)
            approvals = sum(1 for v in votes if v["approve"])
            completed = approvals * 100 >= MIN_QUORUM_PERCENT * participants
            if completed and b.task_state != TaskState.COMPLETED:
                b.task_state = TaskState.COMPLETED
                self.wallet.credit(sol["node"], b.reward,
                                   reason=f"task {task_id} solved")
            b.hash = b.compute_hash()
            b.mine_block(self.difficulty)
            self._rehash_from(b.index + 1)
            self._save()
            return {"ok": True, "approvals": approvals,
                    "participants": participants,
                    "completed": b.task_state == TaskState.COMPLETED}

    def vote_on_task_difficulty(self, task_id: str, difficulty: int,
                                voter: Optional[str] = None) -> Dict[str, Any]:
        voter = voter or self.node_id
        with self._lock:
            b = self._find_task(task_id)
            if b is None:
                return {"ok": False}
            b.difficulty_votes[voter] = int(difficulty)
            votes = sorted(b.difficulty_votes.values())
            b.difficulty = votes[len(votes) // 2]    # median
            b.hash = b.compute_hash()
            b.mine_block(self.difficulty)
            self._rehash_from(b.index + 1)
            self._save()
            return {"ok": True, "difficulty": b.difficulty}

    def tasks(self, state: Optional[str] = None) -> List[Dict[str, Any]]:
        out = []
        seen = set()
        for b in reversed(self.blocks):
            if b.is_task and b.memory_id not in seen:
                seen.add(b.memory_id)
                if state is None or b.task_state == state:
                    out.append(b.to_dict())
        return out

    def _rehash_from(self, start: int) -> None:
        """Task mutations rewrite a block in place; re-link successors so
        validate_chain stays true."""
        for i in range(start, len(self.blocks)):
            self.blocks[i].prev_hash = self.blocks[i - 1].hash
            self.blocks[i].hash = self.blocks[i].compute_hash()
            self.blocks[i].mine_block(self.difficulty)

    # -- queries -------------------------------------------------------------

    def find_memory(self, memory_id: str) -> Optional[Dict[str, Any]]:
        with self._lock:
            for b in reversed(self.blocks):
                if b.memory_id == memory_id:
                    return b.to_dict()
        return None

    def search_memories(self, query: str) -> List[Dict[str, Any]]:
        q = query.lower()
        out = []
        with self._lock:
            for b in self.blocks[1:]:
                hay = json.dumps(b.memory_data).lower()
                if q in hay:
                    out.append(b.to_dict())
        return out

    def stats(self) -> Dict[str, Any]:
        with self._lock:
            return {
                "length": len(self.blocks),
                "tasks": sum(1 for b in self.blocks if b.is_task),
                "nodes": 1 + len(self.nodes),
                "valid": self.validate_chain(),
                "last_hash": self.last_block().hash,
            }

    # -- node registry -------------------------------------------------------

    def register_node(self, address: str) -> None:
        with self._lock:
            if address and address not in self.nodes:
                self.nodes.append(address)

    # -- persistence ---------------------------------------------------------

    def serialize(self) -> List[Dict[str, Any]]:
        with self._lock:
            return [b.to_dict() for b in self.blocks]

    def _save(self) -> None:
        if not self.persist:
            return
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        tmp = self.path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump({"node_id": self.node_id, "chain": self.serialize(),
                       "nodes": self.nodes}, f)
        os.replace(tmp, self.path)

    def _load(self) -> None:
        try:
            with open(self.path, "r", encoding="utf-8") as f:
                data = json.load(f)
            self.blocks = [MemoryBlock.from_dict(d) for d in data.get("chain", [])]
            self.nodes = data.get("nodes", [])
        except (OSError, json.JSONDecodeError, KeyError) as e:
            logger.warning("chain load failed: %s", e)
            self.blocks = []
