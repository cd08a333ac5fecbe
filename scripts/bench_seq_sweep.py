"""Decode tok/s across context lengths and FEI_ATTN_SPLITS (post-nt-fix
re-tune: the weight stream no longer thrashes L2, which can move the
split optimum). One engine per (splits) value; seq swept by prefill."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def measure(splits: int, seqs):
    os.environ["FEI_ATTN_SPLITS"] = str(splits)
    from fei_amd.engine.engine import LocalEngine

    eng = LocalEngine.create("llama3-8b", max_seq_len=8192, seed=7)
    rng = torch.Generator().manual_seed(9)
    out = {}
    for seq in seqs:
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
    eng.shutdown()
    del eng
    torch.cuda.empty_cache()
    return out


if __name__ == "__main__":
    seqs = [512, 2048, 3400, 7900]
    res = {s: measure(s, seqs) for s in (16, 32, 64)}
    print(json.dumps(res))
