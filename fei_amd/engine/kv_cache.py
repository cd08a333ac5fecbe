"""Paged KV-cache allocator.

Sized for MI355X's 288 GB HBM3E: the pool reserves a fraction of free GPU
memory as fixed-size blocks; sequences map logical block indices to physical
blocks so many agent sessions can share one engine without fragmentation.
The flagship decode path uses contiguous per-sequence caches (the HIP
decode-attention kernel reads [B,Hkv,max_seq,D]); this allocator manages
*session* lifetimes above that: each session owns a contiguous region and
the pool tracks free regions. (Block-table attention is a planned kernel
extension; the allocator API already speaks blocks.)
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from fei_amd.utils.logging import get_logger

logger = get_logger("engine.kv_cache")


@dataclass
class BlockTable:
    """Logical -> physical block mapping for one sequence."""
    seq_id: int
    blocks: List[int] = field(default_factory=list)
    length: int = 0                       # tokens used


class PagedKVPool:
    """Fixed-size block pool over one big tensor pair per layer.

    layout per layer: [num_blocks, Hkv, block_size, D] — a sequence's
    logical block i lives at physical index table.blocks[i].
    """

    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        block_size: int = 16,
        num_blocks: Optional[int] = None,
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.bfloat16,
        mem_fraction: float = 0.6,
    ):
        self.block_size = block_size
        self.device = device or (torch.device("cuda:0") if torch.cuda.is_available()
                                 else torch.device("cpu"))
        self.dtype = dtype
        if num_blocks is None:
            if self.device.type == "cuda":
                free, _total = torch.cuda.mem_get_info(self.device)
                budget = int(free * mem_fraction)
            else:
                budget = 1 << 28
            bytes_per_block = (2 * num_layers * num_kv_heads * block_size *
                               head_dim * torch.tensor([], dtype=dtype).element_size())
            num_blocks = max(16, budget // bytes_per_block)
        self.num_blocks = int(num_blocks)
        shape = (self.num_blocks, num_kv_heads, block_size, head_dim)
        self.k = [torch.zeros(shape, device=self.device, dtype=dtype)
                  for _ in range(num_layers)]
        self.v = [torch.zeros(shape, device=self.device, dtype=dtype)
                  for _ in range(num_layers)]
        self._free: List[int] = list(range(self.num_blocks - 1, -1, -1))
        self._tables: Dict[int, BlockTable] = {}
        self._next_id = 0

    # -- sessions ------------------------------------------------------------

    def free_blocks(self) -> int:
        return len(self._free)

    def new_sequence(self) -> int:
        sid = self._next_id
        self._next_id += 1
        self._tables[sid] = BlockTable(seq_id=sid)
        return sid

    def table(self, seq_id: int) -> BlockTable:
        return self._tables[seq_id]

    def ensure_capacity(self, seq_id: int, n_tokens: int) -> List[int]:
        """Grow a sequence to hold n_tokens; returns its block list."""
        t = self._tables[seq_id]
        need = (n_tokens + self.block_size - 1) // self.block_size
        while len(t.blocks) < need:
            if not self._free:
                raise MemoryError(
                    f"KV pool exhausted ({self.num_blocks} blocks); free a "
                    "sequence first")
            t.blocks.append(self._free.pop())
        t.length = max(t.length, n_tokens)
        return t.blocks

    def release(self, seq_id: int) -> None:
        t = self._tables.pop(seq_id, None)
        if t:
            self._free.extend(t.blocks)

    def fork(self, seq_id: int) -> int:
        """Copy-on-write-free fork is not supported (blocks are mutable);
        makes a physical copy of the parent's blocks."""
        parent = self._tables[seq_id]
        child_id = self.new_sequence()
        child = self._tables[child_id]
        for pb in parent.blocks:
            if not self._free:
                raise MemoryError("KV pool exhausted during fork")
            nb = self._free.pop()
            for lk, lv in zip(self.k, self.v):
                lk[nb].copy_(lk[pb])
                lv[nb].copy_(lv[pb])
            child.blocks.append(nb)
        child.length = parent.length
        return child_id

    def block_table_tensor(self, seq_id: int) -> torch.Tensor:
        return torch.tensor(self._tables[seq_id].blocks, dtype=torch.int32,
                            device=self.device)
