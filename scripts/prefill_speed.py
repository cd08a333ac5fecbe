import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from fei_amd.engine.engine import LocalEngine
eng = LocalEngine.create("llama3-8b", max_seq_len=4096, seed=7, use_hip_graph=False)
ids = list(range(4, 2052))
for rep in range(3):
    eng.pos.zero_()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    eng.prefill(ids[:2048])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"prefill 2048: {2048/dt:.0f} tok/s ({dt*1000:.1f} ms)")
