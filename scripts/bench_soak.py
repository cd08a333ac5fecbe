"""Full-context soak: decode from a short prompt to the 8k window on the
shipped defaults; reports mean tok/s over the whole soak + per-quarter
means (decode slows as attention reads more KV)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from fei_amd.engine.engine import LocalEngine

eng = LocalEngine.create("llama3-8b", max_seq_len=8192, seed=7)
eng.prefill(list(range(4, 132)))
quarters = []
t_all = time.perf_counter()
for q in range(4):
    n = 1970
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        eng._graph.replay()
    torch.cuda.synchronize()
    quarters.append(round(n / (time.perf_counter() - t0), 1))
total = 4 * 1970
dt = time.perf_counter() - t_all
print(json.dumps({"soak_tokens": total, "mean_tok_s": round(total / dt, 1),
                  "quarter_tok_s": quarters,
                  "final_pos": int(eng.pos[0]),
                  "fail": int(eng._stream_ws["fail"][0])
                  if eng.stream_decode else 0}))
