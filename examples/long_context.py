"""Long-context decode: size the KV caches toward the 288 GB of HBM3E.

`LocalEngine(max_seq_len="hbm")` probes free device memory after weight
init and sizes the KV window accordingly (8B resolves to the 256k cap,
~34 GB of KV) — SURVEY §5's long-context note. On CPU it falls back to
the model's nominal window; run on an MI355X for the real thing:

    python examples/long_context.py [--model llama3-8b] [--prompt-len 9000]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from fei_amd.engine.engine import LocalEngine


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=None)
    p.add_argument("--prompt-len", type=int, default=None)
    args = p.parse_args()
    gpu = torch.cuda.is_available()
    model = args.model or ("llama3-8b" if gpu else "llama3-tiny")
    n = args.prompt_len or (9000 if gpu else 200)

    eng = LocalEngine.create(model, max_seq_len="hbm", seed=7)
    kv_gb = (eng.k_caches[0].numel() * eng.k_caches[0].element_size()
             * 2 * len(eng.k_caches)) / 2 ** 30
    print(f"window: {eng.max_seq_len} tokens ({kv_gb:.1f} GB KV)")

    rng = torch.Generator().manual_seed(5)
    ids = torch.randint(4, min(16000, eng.spec.vocab_size - 1),
                        (n,), generator=rng).tolist()
    out = eng.generate(ids, max_new_tokens=16, stop_on_eos=False)
    print(f"prefilled {n} tokens at {out['prefill_tok_s']:.0f} tok/s, "
          f"decoded {out['new_tokens']} at {out['decode_tok_s']:.1f} tok/s "
          f"(position now {int(eng.pos[0])})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
