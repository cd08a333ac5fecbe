"""Federation runtime: one agent per GPU rank (BASELINE.json configs[4]).

Each rank owns a LocalEngine on its GPU (or the stub backend on CPU), the
full tool registry with a private memdir, and a memorychain replica.
Shared memories go through ``propose_round`` — a COLLECTIVE consensus
round over RCCL/xGMI (gloo on CPU): broadcast proposal, local votes,
all-gather, deterministic append on quorum. The HTTP node
(fei_amd/memorychain/node.py) remains the wire-compatible control plane
for off-node peers; this runtime is the fast path for the 8 agents of one
MI355X box.

Launch: ``torch.distributed.run --nproc-per-node 8 -m fei_amd.federation``
(one rank per GPU).
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, Optional

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.memorychain.wallet import FeiCoinWallet
from fei_amd.memorychain.xgmi_sync import XgmiSync, propose_collective
from fei_amd.parallel.pg import ParallelContext, barrier, init_from_env
from fei_amd.utils.logging import get_logger

logger = get_logger("federation")


class FederationAgent:
    def __init__(self, ctx: Optional[ParallelContext] = None,
                 provider: str = "local", model: str = "llama3-8b",
                 workdir: Optional[str] = None, engine_kwargs=None):
        self.ctx = ctx or init_from_env()
        rank = self.ctx.rank
        self.node_id = f"agent{rank}"
        workdir = workdir or os.path.join(os.path.expanduser("~"), ".fei",
                                          "federation")
        os.makedirs(workdir, exist_ok=True)
        self.memdir_base = os.path.join(workdir, f"Memdir-{rank}")

        from fei_amd.core.assistant import Assistant
        from fei_amd.tools.code import create_code_tools
        from fei_amd.tools.memory_tools import create_memory_tools
        from fei_amd.tools.registry import ToolRegistry

        registry = ToolRegistry()
        create_code_tools(registry)
        create_memory_tools(registry, base=self.memdir_base)
        kwargs: Dict[str, Any] = {}
        if provider == "local":
            import torch
            device = (torch.device(f"cuda:{self.ctx.local_rank}")
                      if torch.cuda.is_available() else torch.device("cpu"))
            from fei_amd.core.backends import LocalBackend
            from fei_amd.engine.config import get_spec
            from fei_amd.engine.engine import LocalEngine
            if device.type == "cpu" and get_spec(model).hidden_size >= 2048:
                # building 8B+ at fp32 in host RAM takes minutes and tens
                # of GB per rank — same guard as LocalBackend, demo-grade
                # fallback instead of a hang-like init
                logger.warning("no GPU: federation falls back to "
                               "llama3-tiny (asked for %s)", model)
                model = "llama3-tiny"
            engine = LocalEngine.create(model, device=device,
                                        **(engine_kwargs or {}))
            kwargs["backend"] = LocalBackend(engine=engine)
        self.assistant = Assistant(provider=provider, tool_registry=registry,
                                   **kwargs)

        self.chain = MemoryChain(
            node_id=self.node_id,
            path=os.path.join(workdir, f"chain-{rank}.json"),
            difficulty=1,
            wallet=FeiCoinWallet(path=os.path.join(workdir,
                                                   f"wallet-{rank}.json")))
        self.sync = XgmiSync(self.chain, self.ctx)

    # -- collective ops (all ranks call together) ----------------------------

    def propose_round(self, src: int, memory_id: Optional[str] = None,
                      memory_data: Optional[Dict[str, Any]] = None,
                      timestamp: Optional[float] = None) -> Dict[str, Any]:
        """One consensus round; ``src`` supplies the proposal, every rank
        participates. Non-src ranks pass placeholders."""
        if self.ctx.rank == src:
            assert memory_id and memory_data is not None
            timestamp = timestamp or time.time()
        return propose_collective(self.sync, memory_id or "",
                                  memory_data or {}, src,
                                  timestamp or 0.0)

    def share_conversation(self, src: int, subject: str = "") -> Dict[str, Any]:
        """src publishes its current conversation to the chain."""
        memory_id = None
        data = None
        ts = None
        if self.ctx.rank == src:
            lines = []
            for msg in self.assistant.conversation.messages[-10:]:
                content = msg.get("content", "")
                if isinstance(content, list):
                    content = " ".join(
                        str(b.get("text", b.get("content", "")))
                        for b in content if isinstance(b, dict))
                lines.append(f"{msg.get('role')}: {str(content)[:400]}")
            ts = time.time()
            memory_id = f"conv-{self.node_id}-{int(ts * 1000)}"
            data = {"Subject": subject or f"Conversation from {self.node_id}",
                    "Tags": "conversation,federation",
                    "content": "\n".join(lines)}
        return self.propose_round(src, memory_id, data, ts)

    def chain_sync(self) -> bool:
        return self.sync.sync_round()


def main() -> int:
    """Demo loop: every rank answers one prompt and shares it (round-robin
    proposer), then the chains are verified identical."""
    ctx = init_from_env()
    provider = os.environ.get("FED_PROVIDER", "local")
    model = os.environ.get("FED_MODEL", "llama3-8b")
    agent = FederationAgent(ctx, provider=provider, model=model)
    prompt = os.environ.get("FED_PROMPT", "Summarize what this repo does.")
    agent.assistant.ask(prompt)
    for src in range(max(ctx.world_size, 1)):
        out = agent.share_conversation(src)
        if ctx.rank == 0:
            logger.warning("round src=%d accepted=%s votes=%s", src,
                           out.get("accepted"), out.get("votes"))
    barrier(ctx)
    ok = agent.chain.validate_chain()
    print(json.dumps({"rank": ctx.rank, "chain_len": len(agent.chain.blocks),
                      "valid": ok, "last": agent.chain.last_block().hash}))
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
