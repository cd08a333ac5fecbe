"""A/B: persistent weight-streaming decode engine vs the launch path.

Same engine config, same prompt, hipGraph both sides; reports ms/step and
tok/s for each arm plus the ratio. Run on an MI355X:
    timeout 600 python scripts/bench_stream.py [--steps 256] [--seq 512]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run_arm(stream: bool, steps: int, warmup: int, seq: int, model: str):
    if stream:
        os.environ["FEI_STREAM_DECODE"] = "1"
    else:
        os.environ.pop("FEI_STREAM_DECODE", None)
    from fei_amd.engine.engine import LocalEngine

    eng = LocalEngine.create(model, max_seq_len=seq + steps + warmup + 64,
                             seed=7, use_hip_graph=True)
    rng = torch.Generator().manual_seed(99)
    prompt = torch.randint(4, 260, (seq,), generator=rng).tolist()
    eng.ensure_graph()
    eng.prefill(prompt)
    for _ in range(warmup):
        eng._graph.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng._graph.replay()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if stream:
        assert int(eng._stream_ws["fail"][0]) == 0, "stream engine gave up"
        assert eng.stream_decode, "stream engine did not enable"
    first_toks = eng.out_tokens[0, :8].tolist()
    eng.shutdown()
    del eng
    torch.cuda.empty_cache()
    return {"stream": stream, "ms_per_step": round(dt / steps * 1000, 4),
            "tok_s": round(steps / dt, 2), "first_toks": first_toks}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=256)
    p.add_argument("--warmup", type=int, default=32)
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--stream-only", action="store_true")
    args = p.parse_args()
    arms = [True] if args.stream_only else [False, True]
    res = [run_arm(s, args.steps, args.warmup, args.seq, args.model)
           for s in arms]
    out = {"model": args.model, "seq": args.seq, "steps": args.steps,
           "arms": res}
    if len(res) == 2:
        out["ratio_stream_vs_launch"] = round(
            res[1]["ms_per_step"] / res[0]["ms_per_step"], 4)
        out["tokens_agree"] = res[0]["first_toks"] == res[1]["first_toks"]
    print(json.dumps(out))


if __name__ == "__main__":
    main()
