"""Multi-session continuous batching over the paged KV pool.
CPU demo uses llama3-tiny; on an MI355X switch to llama3-8b."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from fei_amd.engine.engine import LocalEngine
from fei_amd.engine.sessions import PagedSessionManager

model = "llama3-8b" if torch.cuda.is_available() else "llama3-tiny"
engine = LocalEngine.create(model)
mgr = PagedSessionManager(engine, block_size=16, num_blocks=256)

sids = [mgr.open(p, max_new_tokens=32) for p in
        ["def quicksort(xs):", "SELECT name FROM users WHERE", "# TODO:"]]
mgr.step(); mgr.step()
late = mgr.open("print('joined mid-flight')", max_new_tokens=16)
mgr.run()

for sid in sids + [late]:
    res = mgr.result(sid)
    print(f"[{sid}] done={res['done']} {res['text'][:60]!r}")
    mgr.close(sid)
print("free blocks:", mgr.pool.free_blocks())
