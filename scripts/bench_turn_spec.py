"""A/B: tool-turn p50 with prompt-lookup speculative decoding on vs off.

Speculative decode is token-identical at greedy (CPU contract tests pin
it); agent turns decode text that repeats the context, its best case.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run(spec: bool):
    os.environ["FEI_SPEC_DECODE"] = "1" if spec else "0"
    from fei_amd.core.turn_bench import measure_tool_turns
    from fei_amd.engine.engine import LocalEngine

    eng = LocalEngine.create("llama3-8b", max_seq_len=2048, seed=7)
    out = measure_tool_turns(eng, n_turns=9, max_new=96)
    eng.shutdown()
    del eng
    import torch
    torch.cuda.empty_cache()
    return out


if __name__ == "__main__":
    off = run(False)
    on = run(True)
    print(json.dumps({"spec_off_p50_s": off["tool_turn_p50_s"],
                      "spec_on_p50_s": on["tool_turn_p50_s"],
                      "ratio": round(on["tool_turn_p50_s"]
                                     / off["tool_turn_p50_s"], 3)}))
