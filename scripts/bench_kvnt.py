"""A/B the decode-attention KV cache-policy (FEI_ATTN_KV_NT) across
context lengths: nt bypasses L1/L2 on K/V reads — expected to lose when
the KV fits L2 (short ctx) and win once it streams from HBM (long ctx).
One process per setting (the C launcher caches the env once)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def main():
    from fei_amd.engine.engine import LocalEngine

    eng = LocalEngine.create("llama3-8b", max_seq_len=8192, seed=7)
    rng = torch.Generator().manual_seed(9)
    out = {}
    for seq in (512, 2048, 4096, 7900):
        ids = torch.randint(4, 16000, (seq,), generator=rng).tolist()
        eng.prefill(ids)
        for _ in range(24):
            eng._graph.replay()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(128):
            eng._graph.replay()
        torch.cuda.synchronize()
        out[seq] = round(128 / (time.perf_counter() - t0), 1)
    print(json.dumps({"kv_nt": os.environ.get("FEI_ATTN_KV_NT", "0"),
                      "tok_s": out}))


if __name__ == "__main__":
    main()
