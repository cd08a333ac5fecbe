"""Bulk memorychain sync over RCCL/xGMI.

The reference gossips the FULL serialized chain as JSON over HTTP on every
update — O(chain) bytes per peer per proposal. For the
8-agents-on-one-node federation (BASELINE.json configs[4]) the control
plane (votes, registration — small messages) stays HTTP wire-compatible,
while BULK payloads (chain sync, embedding-index shards) move over
torch.distributed collectives: RCCL over xGMI when each agent owns a GPU,
gloo on CPU (which is how the multi-process test runs it).

Protocol (anti-entropy round, collective and deterministic):
  1. all_gather of (chain_length, crc32) from every rank
  2. the best rank (longest chain; rank id breaks ties) broadcasts its
     serialized chain as a uint8 tensor (length first, then payload)
  3. every other rank applies it through the normal
     ``MemoryChain.receive_chain_update`` validation (prefix rule intact)

xGMI note: broadcast from one source is per-link bound on the ring; at
8 ranks and chain sizes ≤ a few MB this is latency-dominated and still
~100x the HTTP/JSON path of the reference.
"""

from __future__ import annotations

import json
import zlib
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.parallel.pg import ParallelContext, get_world
from fei_amd.utils.logging import get_logger

logger = get_logger("memorychain.xgmi_sync")


def _comm_device(ctx: ParallelContext) -> torch.device:
    if ctx.backend == "nccl":
        return torch.device("cuda", ctx.local_rank)
    return torch.device("cpu")


def broadcast_bytes(data: Optional[bytes], src: int,
                    ctx: Optional[ParallelContext] = None) -> bytes:
    """Collective: src's bytes end up on every rank."""
    ctx = ctx or get_world()
    dev = _comm_device(ctx)
    if ctx.rank == src:
        assert data is not None
        length = torch.tensor([len(data)], dtype=torch.int64, device=dev)
    else:
        length = torch.zeros(1, dtype=torch.int64, device=dev)
    dist.broadcast(length, src=src, group=ctx.group)
    n = int(length.item())
    if ctx.rank == src:
        buf = torch.frombuffer(bytearray(data), dtype=torch.uint8).to(dev)
    else:
        buf = torch.empty(n, dtype=torch.uint8, device=dev)
    dist.broadcast(buf, src=src, group=ctx.group)
    return bytes(buf.cpu().numpy().tobytes())


def broadcast_tensor(t: Optional[torch.Tensor], src: int, shape, dtype,
                     ctx: Optional[ParallelContext] = None) -> torch.Tensor:
    """Collective tensor broadcast (embedding-index shards)."""
    ctx = ctx or get_world()
    dev = _comm_device(ctx)
    if ctx.rank == src:
        buf = t.to(dev)
    else:
        buf = torch.empty(*shape, dtype=dtype, device=dev)
    dist.broadcast(buf, src=src, group=ctx.group)
    return buf


class XgmiSync:
    """Anti-entropy chain synchronisation for co-located federation ranks."""

    def __init__(self, chain: MemoryChain, ctx: Optional[ParallelContext] = None):
        self.chain = chain
        self.ctx = ctx or get_world()

    def _local_state(self) -> Tuple[int, int]:
        blob = json.dumps(self.chain.serialize(), sort_keys=True).encode()
        return len(self.chain.blocks), zlib.crc32(blob)

    def sync_round(self) -> bool:
        """One collective round; returns True if this rank's chain changed.
        Every rank in the group MUST call this together."""
        ctx = self.ctx
        if not ctx.is_distributed:
            return False
        dev = _comm_device(ctx)
        length, crc = self._local_state()
        state = torch.tensor([length, crc], dtype=torch.int64, device=dev)
        states: List[torch.Tensor] = [torch.zeros(2, dtype=torch.int64, device=dev)
                                      for _ in range(ctx.world_size)]
        dist.all_gather(states, state, group=ctx.group)
        lengths = [int(s[0]) for s in states]
        crcs = [int(s[1]) for s in states]
        best = max(range(ctx.world_size), key=lambda r: (lengths[r], -r))
        if lengths.count(lengths[best]) == ctx.world_size and \
                len(set(crcs)) == 1:
            return False                                    # already in sync
        payload = None
        if ctx.rank == best:
            payload = json.dumps(self.chain.serialize()).encode()
        blob = broadcast_bytes(payload, src=best, ctx=ctx)
        if ctx.rank == best:
            return False
        try:
            blocks = json.loads(blob.decode())
        except json.JSONDecodeError:
            logger.warning("bad chain payload in sync_round")
            return False
        return self.chain.receive_chain_update(blocks)

    def sync_embeddings(self, embeddings: Optional[torch.Tensor],
                        src: int) -> torch.Tensor:
        """Share an embedding-index shard from ``src`` with every rank."""
        ctx = self.ctx
        if not ctx.is_distributed:
            return embeddings
        if ctx.rank == src:
            shape = torch.tensor(list(embeddings.shape), dtype=torch.int64)
        else:
            shape = torch.zeros(2, dtype=torch.int64)
        dev = _comm_device(ctx)
        shape = shape.to(dev)
        dist.broadcast(shape, src=src, group=ctx.group)
        return broadcast_tensor(embeddings, src, tuple(int(x) for x in shape),
                                torch.float16 if embeddings is None
                                else embeddings.dtype, ctx=ctx)


def propose_collective(sync: "XgmiSync", memory_id: str, memory_data: dict,
                       src: int, timestamp: float) -> dict:
    """Collective consensus round for co-located federation ranks: the
    proposal is broadcast from ``src``, every rank votes locally, votes are
    all-gathered, and on quorum EVERY rank appends the identical block
    (fixed timestamp + deterministic mining) — no chain gossip, no
    divergence window, unlike the reference's racy HTTP broadcast
    (SURVEY.md §7 hard-part 5). All ranks MUST call together.
    """
    from fei_amd.memorychain.chain import MIN_QUORUM_PERCENT

    ctx = sync.ctx
    chain = sync.chain
    if not ctx.is_distributed:
        return chain.propose_memory(memory_id, memory_data)
    payload = None
    if ctx.rank == src:
        payload = json.dumps({"memory_id": memory_id,
                              "memory_data": memory_data,
                              "proposer_node": chain.node_id,
                              "timestamp": timestamp}).encode()
    blob = broadcast_bytes(payload, src=src, ctx=ctx)
    proposal = json.loads(blob.decode())
    my_vote = 1 if chain.vote_on_proposal(proposal) else 0
    dev = _comm_device(ctx)
    vote = torch.tensor([my_vote], dtype=torch.int64, device=dev)
    votes = [torch.zeros(1, dtype=torch.int64, device=dev)
             for _ in range(ctx.world_size)]
    dist.all_gather(votes, vote, group=ctx.group)
    total_votes = sum(int(v) for v in votes)
    accepted = total_votes * 100 >= MIN_QUORUM_PERCENT * ctx.world_size
    out = {"accepted": accepted, "votes": total_votes,
           "total": ctx.world_size}
    if accepted:
        block = chain.add_memory(proposal["memory_id"],
                                 proposal["memory_data"],
                                 proposer=proposal["proposer_node"],
                                 timestamp=proposal["timestamp"])
        out["block_index"] = block.index
        out["block_hash"] = block.hash
    return out
