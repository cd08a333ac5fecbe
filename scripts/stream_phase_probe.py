"""Phase breakdown of the persistent stream layer: s_memtime stamps per WG.

Runs ONE layer repeatedly with the dbg buffer enabled and prints the mean
per-phase deltas (in us at 2.0 GHz nominal shader clock) over all 256 WGs.
"""
import os, sys, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fei_amd import ops
from fei_amd.engine.engine import LocalEngine

eng = LocalEngine.create("llama3-8b", max_seq_len=1024, seed=7,
                         use_hip_graph=False)
eng.prefill(eng.tokenizer.encode("probe " * 100))
spec, model = eng.spec, eng.model
ws = ops.stream_workspace(spec, eng.device)
ws["dbg"] = torch.zeros(256 * 16, dtype=torch.int64, device=eng.device)
bufs = (torch.zeros(1, spec.hidden_size, dtype=eng.dtype, device=eng.device),
        torch.zeros(1, spec.hidden_size, dtype=eng.dtype, device=eng.device))
# warm
for _ in range(3):
    model.forward_decode_stream(eng.token, eng.pos, eng.k_caches,
                                eng.v_caches, ws, bufs)
    torch.cuda.synchronize()
    eng.pos += 1
assert int(ws["fail"][0]) == 0
d = ws["dbg"].view(256, 16).cpu().numpy().astype("uint64")
# stamps: 0 start, 1 x1ready, 2 s1done, 3 partials, 4 attgathered,
# 5 s3done, 6 x2ready, 7 s4done, 8 actgathered, 9 s5done, 14 loaderstart,
# 15 loaderend
names = ["start", "x1_ready", "s1_slots", "s2_partials", "att_gather",
         "s3_slots", "x2_ready", "s4_slots", "act_gather", "s5_done"]
t0 = d[:, 0].astype("int64")
clk = 2.0e3  # cycles per us (approx; relative shares are what matter)
out = {}
prev = t0
for i, nm in enumerate(names[1:], start=1):
    cur = d[:, i].astype("int64")
    if i in (2, 3, 4):         # attention stamp chain: WGs 0..7 only
        dt = (cur[:8] - prev[:8]) / clk
    else:
        dt = (cur - prev) / clk
    out[nm] = round(float(dt.mean()), 2)
    prev = cur
out["wave0_total"] = round(float(((d[:, 9].astype("int64") - t0) / clk).mean()), 2)
out["loader_total"] = round(float(((d[:, 15].astype("int64") - d[:, 14].astype("int64")) / clk).mean()), 2)
ld = d[:, [14, 10, 11, 12, 13]].astype("int64")
for nm, a_, b_, ns in (("ld_s1", 0, 1, 12), ("ld_s3", 1, 2, 8),
                       ("ld_s4", 2, 3, 56), ("ld_s5", 3, 4, 32)):
    out[nm + "_per_slot"] = round(float(((ld[:, b_] - ld[:, a_]) / clk).mean())
                                  / ns, 3)
print(json.dumps(out))
