"""Prompt-lookup speculative decoding: the context drafts, one prefill
verifies. Greedy output; acceptance metrics in the result."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from fei_amd.engine.engine import LocalEngine

model = "llama3-8b" if torch.cuda.is_available() else "llama3-tiny"
engine = LocalEngine.create(model)
prompt = ("for i in range(10):\n    print(i)\n" * 8 +
          "for i in range(10):\n")
out = engine.generate(prompt, max_new_tokens=64, speculative=True)
print(out["text"][:120])
print({k: v for k, v in out.items() if str(k).startswith("spec_")})
