"""Direct engine usage (GPU): Llama-3-8B generation with metrics.
Run on an MI355X: python examples/engine_generate.py"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from fei_amd.engine.engine import LocalEngine

model = "llama3-8b" if torch.cuda.is_available() else "llama3-tiny"
engine = LocalEngine.create(model, max_seq_len=1024)
out = engine.generate("def fibonacci(n):", max_new_tokens=64, stop_on_eos=False)
print(out["text"][:200])
print({k: round(v, 2) if isinstance(v, float) else v
       for k, v in engine.last_metrics.items()})
