"""Generate sample memories for demos/tests
(reference parity: memdir_tools/create_samples.py)."""

from __future__ import annotations

import random
from typing import Optional

from fei_amd.memdir import utils as mu

_SUBJECTS = [
    "Fix the flaky scheduler test", "Refactor the KV-cache allocator",
    "Benchmark decode attention", "Write rocprof analysis notes",
    "Investigate RCCL bucket sizing", "Update the repo map generator",
    "Profile RMSNorm kernel", "Plan the memorychain demo",
    "Review tokenizer edge cases", "Clean up backup directories",
]
_TAGS = ["project", "gpu", "kernels", "memory", "agent", "todo", "perf", "docs"]
_BODIES = [
    "Notes from today's debugging session.\nThe root cause was a stale cache entry.",
    "Performance numbers look promising; need a second run to confirm.",
    "TODO: split this into smaller pieces and add tests.",
    "This is done and verified on the target hardware.",
    "Important: keep the on-disk format byte-compatible.",
]


def create_samples(count: int = 25, base: Optional[str] = None, seed: int = 0) -> int:
    rng = random.Random(seed)
    folders = ["", ".Projects", ".ToDoLater"]
    for _ in range(count):
        headers = {
            "Subject": rng.choice(_SUBJECTS),
            "Tags": ",".join(rng.sample(_TAGS, k=rng.randint(1, 3))),
            "Priority": rng.choice(["low", "normal", "high"]),
        }
        flags = "".join(rng.sample("SRFP", k=rng.randint(0, 2)))
        mu.create_memory(rng.choice(folders), headers, rng.choice(_BODIES),
                         flags=flags, base=base, status=rng.choice(["cur", "new"]))
    return count
