"""Model specs for the local engine.

Llama-3 8B/70B shapes per the BASELINE.json configs (weights are random-init
— there is no network for checkpoints; bench.py states data=synthetic).
``llama3-tiny`` exists for CPU tests; ``bge-base`` shapes the memdir
embedding encoder (fei_amd/models/bge.py).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict


@dataclass(frozen=True)
class ModelSpec:
    name: str
    vocab_size: int
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    intermediate_size: int
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    max_seq_len: int = 8192
    arch: str = "llama"           # "llama" (decoder) | "bert" (encoder)

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def params_bytes(self, dtype_bytes: int = 2) -> int:
        """Approximate parameter bytes (embeddings + blocks + head)."""
        h, i, v = self.hidden_size, self.intermediate_size, self.vocab_size
        per_layer = (h * (self.q_size + 2 * self.kv_size)   # qkv
                     + self.q_size * h                      # o
                     + 3 * h * i                            # gate/up/down
                     + 2 * h)                               # norms
        total = v * h * 2 + self.num_layers * per_layer + h
        return total * dtype_bytes


MODEL_SPECS: Dict[str, ModelSpec] = {
    "llama3-8b": ModelSpec(
        name="llama3-8b", vocab_size=128256, hidden_size=4096, num_layers=32,
        num_heads=32, num_kv_heads=8, head_dim=128, intermediate_size=14336,
    ),
    "llama3-70b": ModelSpec(
        name="llama3-70b", vocab_size=128256, hidden_size=8192, num_layers=80,
        num_heads=64, num_kv_heads=8, head_dim=128, intermediate_size=28672,
    ),
    # 1B-ish shape for quick single-GPU smoke/perf probes
    "llama3-1b": ModelSpec(
        name="llama3-1b", vocab_size=128256, hidden_size=2048, num_layers=16,
        num_heads=32, num_kv_heads=8, head_dim=64, intermediate_size=8192,
    ),
    # tiny: CPU tests and GPU numerics tests
    "llama3-tiny": ModelSpec(
        name="llama3-tiny", vocab_size=512, hidden_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, head_dim=64, intermediate_size=512,
        max_seq_len=512,
    ),
    # tiny8: every sharded axis divisible by 8 — the gloo-8 TP harness
    # (70B's tp=8 shard shape ratios at toy scale)
    "llama3-tiny8": ModelSpec(
        name="llama3-tiny8", vocab_size=512, hidden_size=256, num_layers=2,
        num_heads=8, num_kv_heads=8, head_dim=32, intermediate_size=512,
        max_seq_len=512,
    ),
    # embedding encoder shape (bge-base class): 12L/768h/12heads, D=64
    "bge-base": ModelSpec(
        name="bge-base", vocab_size=30522, hidden_size=768, num_layers=12,
        num_heads=12, num_kv_heads=12, head_dim=64, intermediate_size=3072,
        max_seq_len=512, arch="bert",
    ),
}


def get_spec(name: str) -> ModelSpec:
    if name not in MODEL_SPECS:
        raise KeyError(f"unknown model {name!r}; known: {sorted(MODEL_SPECS)}")
    return MODEL_SPECS[name]
