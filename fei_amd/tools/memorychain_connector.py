"""HTTP client for a memorychain node.

Parity: reference MemorychainConnector (fei/tools/memorychain_connector.py:
33-643): env ``MEMORYCHAIN_NODE`` default localhost:6789, add_memory ->
POST /propose, chain fetch, client-side linear search over the full chain,
chain stats, ``#mem:id`` reference extraction/resolution, validate_chain
with a local rebuild fallback, status APIs, and the conversation helper.
"""

from __future__ import annotations

import os
import re
from typing import Any, Dict, List, Optional

import requests

from fei_amd.utils.logging import get_logger

logger = get_logger("tools.memorychain_connector")

MEM_REF_RE = re.compile(r"#mem:([0-9a-zA-Z_-]+)")


class MemorychainConnector:
    def __init__(self, node: Optional[str] = None, timeout: float = 10.0):
        self.node = node or os.environ.get("MEMORYCHAIN_NODE", "localhost:6789")
        self.timeout = timeout

    def _url(self, path: str) -> str:
        return f"http://{self.node}/memorychain/{path.lstrip('/')}"

    def _get(self, path: str, **params) -> Dict[str, Any]:
        try:
            r = requests.get(self._url(path), params=params or None,
                             timeout=self.timeout)
            return r.json()
        except (requests.RequestException, ValueError) as e:
            return {"error": str(e)}

    def _post(self, path: str, payload: Dict[str, Any]) -> Dict[str, Any]:
        try:
            r = requests.post(self._url(path), json=payload,
                              timeout=self.timeout)
            return r.json()
        except (requests.RequestException, ValueError) as e:
            return {"error": str(e)}

    # -- memories ------------------------------------------------------------

    def is_available(self) -> bool:
        return self._get("health").get("status") == "ok"

    def add_memory(self, headers: Dict[str, str], body: str = "",
                   memory_id: Optional[str] = None) -> Dict[str, Any]:
        data = dict(headers)
        if body:
            data["content"] = body
        payload: Dict[str, Any] = {"memory_data": data}
        if memory_id:
            payload["memory_id"] = memory_id
        return self._post("propose", payload)

    def get_chain(self) -> List[Dict[str, Any]]:
        return self._get("chain").get("chain", [])

    def search_memories(self, query: str) -> List[Dict[str, Any]]:
        """Client-side linear scan over the chain
        (reference: memorychain_connector.py:273-394)."""
        q = query.lower()
        out = []
        for block in self.get_chain()[1:]:
            import json as _json
            if q in _json.dumps(block.get("memory_data", {})).lower():
                out.append(block)
        return out

    def search_by_tag(self, tag: str) -> List[Dict[str, Any]]:
        tag = tag.lower()
        out = []
        for block in self.get_chain()[1:]:
            tags = str(block.get("memory_data", {}).get("Tags", "")).lower()
            if tag in [t.strip() for t in re.split(r"[,\s]+", tags) if t]:
                out.append(block)
        return out

    def get_memories_with_status(self, status: str) -> List[Dict[str, Any]]:
        return [b for b in self.get_chain()[1:]
                if b.get("memory_data", {}).get("Status", "").lower() == status.lower()]

    def get_chain_stats(self) -> Dict[str, Any]:
        chain = self.get_chain()
        tasks = [b for b in chain if b.get("is_task")]
        return {
            "length": len(chain),
            "tasks": len(tasks),
            "completed_tasks": sum(1 for t in tasks
                                   if t.get("task_state") == "completed"),
            "proposers": sorted({b.get("proposer_node", "") for b in chain[1:]}),
        }

    # -- #mem: references ----------------------------------------------------

    @staticmethod
    def extract_memory_references(text: str) -> List[str]:
        return MEM_REF_RE.findall(text)

    def resolve_memory_references(self, text: str) -> Dict[str, Any]:
        resolved = {}
        chain = self.get_chain()
        for ref in self.extract_memory_references(text):
            for block in chain:
                if block.get("memory_id") == ref or \
                        block.get("memory_id", "").startswith(ref):
                    resolved[ref] = block.get("memory_data", {})
                    break
        return resolved

    # -- validation / status -------------------------------------------------

    def validate_chain(self) -> Dict[str, Any]:
        """Ask the node; fall back to a local rebuild check
        (reference: memorychain_connector.py:543-576)."""
        chain = self.get_chain()
        if not chain:
            return {"valid": False, "reason": "empty or unreachable"}
        from fei_amd.memorychain.chain import MemoryBlock
        blocks = [MemoryBlock.from_dict(d) for d in chain]
        for i, b in enumerate(blocks):
            if b.hash != b.compute_hash():
                return {"valid": False, "reason": f"bad hash at {i}"}
            if i and b.prev_hash != blocks[i - 1].hash:
                return {"valid": False, "reason": f"broken link at {i}"}
        return {"valid": True, "length": len(blocks)}

    def update_status(self, status: str, load: float = 0.0,
                      current_task: Optional[str] = None) -> Dict[str, Any]:
        return self._post("update_status", {
            "status": status, "load": load, "current_task": current_task})

    def get_network_status(self) -> Dict[str, Any]:
        return self._get("network_status")

    # -- tasks ---------------------------------------------------------------

    def propose_task(self, task_data: Dict[str, Any], reward: float = 1.0,
                     difficulty: int = 1) -> Dict[str, Any]:
        return self._post("propose_task", {"task_data": task_data,
                                           "reward": reward,
                                           "difficulty": difficulty})

    def claim_task(self, task_id: str) -> Dict[str, Any]:
        return self._post("claim_task", {"task_id": task_id})

    def submit_solution(self, task_id: str, solution: str) -> Dict[str, Any]:
        return self._post("submit_solution", {"task_id": task_id,
                                              "solution": solution})

    def list_tasks(self, state: Optional[str] = None) -> List[Dict[str, Any]]:
        params = {"state": state} if state else {}
        return self._get("tasks", **params).get("tasks", [])


def add_memory_from_conversation(connector: MemorychainConnector,
                                 messages: List[Dict[str, Any]],
                                 subject: str,
                                 tags: str = "conversation") -> Dict[str, Any]:
    """Summarize a conversation into one chain memory
    (reference helper: memorychain_connector.py:592-643)."""
    lines = []
    for msg in messages[-20:]:
        role = msg.get("role", "?")
        content = msg.get("content", "")
        if isinstance(content, list):
            content = " ".join(str(b.get("text", b.get("content", "")))
                               for b in content if isinstance(b, dict))
        lines.append(f"{role}: {str(content)[:500]}")
    return connector.add_memory(
        {"Subject": subject, "Tags": tags}, body="\n".join(lines))
