"""Continuous-batching serving throughput: N concurrent sessions with
staggered prompt lengths through PagedSessionManager (block-table
attention, per-step admission/retire) — the serving-path counterpart of
bench.py --batch (which measures the static-batch engine loop)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from fei_amd.engine.engine import LocalEngine
from fei_amd.engine.sessions import PagedSessionManager

N = int(os.environ.get("SESS_N", "8"))
NEW = int(os.environ.get("SESS_NEW", "256"))
MODEL = os.environ.get("SESS_MODEL", "llama3-8b")
eng = LocalEngine.create(MODEL, max_seq_len=4096, seed=7)
mgr = PagedSessionManager(eng)
rng = torch.Generator().manual_seed(5)
sids = []
for i in range(N):
    plen = 64 + 48 * i                      # staggered contexts
    hi = min(16000, eng.spec.vocab_size - 2)
    ids = torch.randint(4, hi, (plen,), generator=rng).tolist()
    sids.append(mgr.open(ids, max_new_tokens=NEW))
if torch.cuda.is_available():
    torch.cuda.synchronize()
t0 = time.perf_counter()
steps = 0
CH = int(os.environ.get("SESS_CHUNK", "8"))
while mgr.active and steps < NEW + 8:
    k = min(CH, NEW + 8 - steps)
    mgr.step_chunk(k)
    steps += k
if torch.cuda.is_available():
    torch.cuda.synchronize()
dt = time.perf_counter() - t0
toks = sum(len(mgr.result(s)["token_ids"]) for s in sids)
print(json.dumps({
    "metric": "serving aggregate tok/s (continuous batching, paged KV)",
    "value": round(toks / dt, 1), "unit": "tok/s", "sessions": N,
    "new_tokens_per_session": NEW, "steps": steps,
    "ms_per_step": round(dt / steps * 1000, 3), "model": MODEL,
    "data": "synthetic"}))
