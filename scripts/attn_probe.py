"""Isolate k_attn_decode at a fixed long context: wall time per call ->
achieved KV bandwidth, for rocprofv3 --stats / --pmc attribution."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fei_amd import ops

B, Hq, Hkv, D, MS = 1, 32, 8, 128, 8192
n = int(sys.argv[1]) if len(sys.argv) > 1 else 8190
splits = int(sys.argv[2]) if len(sys.argv) > 2 else 32
dev = "cuda:0"
g = torch.Generator(device=dev).manual_seed(1)
q = (torch.randn(B, Hq, D, generator=g, device=dev) * 0.1).bfloat16()
kc = (torch.randn(B, Hkv, MS, D, generator=g, device=dev) * 0.1).bfloat16()
vc = (torch.randn(B, Hkv, MS, D, generator=g, device=dev) * 0.1).bfloat16()
pos = torch.tensor([n - 1], dtype=torch.int32, device=dev)
ws = (torch.zeros(B, Hq, splits, D, dtype=torch.float32, device=dev),
      torch.zeros(B, Hq, splits, 2, dtype=torch.float32, device=dev))
out = torch.empty_like(q)
for _ in range(20):
    ops.attn_decode(q, kc, vc, pos, splits=splits, workspace=ws, out=out)
torch.cuda.synchronize()
t0 = time.perf_counter()
REP = 300
for _ in range(REP):
    ops.attn_decode(q, kc, vc, pos, splits=splits, workspace=ws, out=out)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / REP
kv_bytes = n * Hkv * D * 2 * 2          # K+V bf16
print(f"n={n} splits={splits}: {dt*1e6:.2f} us/call (attn+combine), "
      f"KV {kv_bytes/1e6:.1f} MB -> {kv_bytes/dt/1e12:.2f} TB/s")
