"""End-to-end federation runtime: 2 agent ranks over gloo, stub backend.
Every rank runs a real agent turn, shares it through the COLLECTIVE
consensus round, and ends with byte-identical valid chains."""

import json
import os
import subprocess
import sys

WORKER = r"""
import json, os
os.environ["FED_PROVIDER"] = "stub"
from fei_amd.federation import FederationAgent
from fei_amd.parallel.pg import barrier, init_from_env

ctx = init_from_env(backend="gloo")
agent = FederationAgent(ctx, provider="stub",
                        workdir=os.environ["FED_WORK"] + f"/r{ctx.rank}")
agent.assistant.ask(f"note from rank {ctx.rank}")

results = []
for src in range(ctx.world_size):
    out = agent.share_conversation(src, subject=f"turn of rank {src}")
    results.append(out)

# one extra round that must be rejected (duplicate id)
dup = agent.propose_round(0, "dup-id", {"Subject": "first"},
                          1000.0) if ctx.rank == 0 else \
      agent.propose_round(0)
dup2 = agent.propose_round(0, "dup-id", {"Subject": "again"},
                           1001.0) if ctx.rank == 0 else \
       agent.propose_round(0)

barrier(ctx)
out = {
    "rank": ctx.rank,
    "accepted": [r["accepted"] for r in results],
    "chain_len": len(agent.chain.blocks),
    "valid": agent.chain.validate_chain(),
    "last_hash": agent.chain.last_block().hash,
    "dup_first": dup["accepted"],
    "dup_second": dup2["accepted"],
    "subjects": [b.memory_data.get("Subject") for b in agent.chain.blocks[1:]],
}
with open(os.path.join(os.environ["FED_WORK"], f"out{ctx.rank}.json"), "w") as f:
    json.dump(out, f)
"""


def test_two_rank_federation(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["FED_WORK"] = str(tmp_path)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.setdefault("PYTHONPATH", os.getcwd())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29537", str(script)],
        env=env, capture_output=True, text=True, timeout=300,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    out0 = json.loads((tmp_path / "out0.json").read_text())
    out1 = json.loads((tmp_path / "out1.json").read_text())
    assert out0["accepted"] == [True, True]
    assert out0["chain_len"] == out1["chain_len"] == 4   # genesis + 2 conv + dup
    assert out0["valid"] and out1["valid"]
    assert out0["last_hash"] == out1["last_hash"]        # identical chains
    assert out0["dup_first"] is True
    assert out0["dup_second"] is False                   # dedupe via votes
    assert out0["subjects"] == out1["subjects"]
