"""Federation bulk-sync over torch.distributed: 2-process gloo test of the
anti-entropy round (the same code path runs RCCL/xGMI on the GPU node)."""

import json
import os
import subprocess
import sys

WORKER = r"""
import json, os, torch
from fei_amd.memorychain.chain import MemoryChain
from fei_amd.memorychain.wallet import FeiCoinWallet
from fei_amd.memorychain.xgmi_sync import XgmiSync, broadcast_bytes
from fei_amd.parallel.pg import init_from_env

ctx = init_from_env(backend="gloo")
base = os.environ["SYNC_TMP"]
chain = MemoryChain(node_id=f"node{ctx.rank}",
                    path=os.path.join(base, f"c{ctx.rank}.json"),
                    difficulty=1,
                    wallet=FeiCoinWallet(path=os.path.join(base, f"w{ctx.rank}.json")))

# rank 0 appends two memories locally; rank 1 stays at genesis
if ctx.rank == 0:
    chain.add_memory("m1", {"Subject": "first"})
    chain.add_memory("m2", {"Subject": "second"})

sync = XgmiSync(chain, ctx)
changed = sync.sync_round()
# second round: everyone equal -> no-op
changed2 = sync.sync_round()

# raw byte broadcast check
blob = broadcast_bytes(b"hello-xgmi" if ctx.rank == 0 else None, src=0, ctx=ctx)

out = {"rank": ctx.rank, "changed": changed, "changed2": changed2,
       "chain_len": len(chain.blocks),
       "subjects": [b.memory_data.get("Subject") for b in chain.blocks[1:]],
       "valid": chain.validate_chain(), "blob": blob.decode()}
with open(os.path.join(base, f"out{ctx.rank}.json"), "w") as f:
    json.dump(out, f)
"""


def test_gloo_chain_sync(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["SYNC_TMP"] = str(tmp_path)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.setdefault("PYTHONPATH", os.getcwd())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", str(script)],
        env=env, capture_output=True, text=True, timeout=240,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    out0 = json.loads((tmp_path / "out0.json").read_text())
    out1 = json.loads((tmp_path / "out1.json").read_text())
    assert out0["chain_len"] == 3 and out1["chain_len"] == 3
    assert out1["changed"] is True         # rank 1 adopted rank 0's chain
    assert out0["changed"] is False        # source keeps its chain
    assert out0["changed2"] is False and out1["changed2"] is False
    assert out1["subjects"] == ["first", "second"]
    assert out1["valid"] is True
    assert out0["blob"] == out1["blob"] == "hello-xgmi"
