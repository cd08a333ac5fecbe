"""p50 tool-turn latency — the second half of the BASELINE metric.

Measures the FULL agent turn through the real stack on the local engine:
Assistant.chat -> prompt render -> prefill -> hipGraph decode until the
model stops (or token cap) -> tool-call parse -> GlobTool execution ->
continuation prefill+decode. Weights are random-init, so the model cannot
emit valid tool calls itself; the turn structure is made deterministic by
injecting the tool round (the scripted-turn harness SURVEY.md §7 calls
for), while ALL latency comes from real engine + tool work.
"""
import json, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    from fei_amd.core.assistant import Assistant
    from fei_amd.core.turn_bench import TurnBackend
    from fei_amd.engine.engine import LocalEngine
    from fei_amd.tools.code import create_code_tools
    from fei_amd.tools.registry import ToolRegistry

    has_gpu = torch.cuda.is_available()
    model = os.environ.get("TURN_MODEL",
                           "llama3-8b" if has_gpu else "llama3-tiny")
    max_new = 128 if has_gpu else 16
    engine = LocalEngine.create(model, max_seq_len=8192 if has_gpu else 512)

    registry = ToolRegistry()
    create_code_tools(registry)
    assistant = Assistant(provider="local", tool_registry=registry,
                          backend=TurnBackend(engine, max_new))

    n_turns = int(os.environ.get("TURN_N", "20" if has_gpu else "3"))
    backend = assistant.providers.get_backend()
    for fixed_budget in (False, True):
        # fixed_budget=True decodes the full token budget every round
        # (random-init weights hit EOS at arbitrary points otherwise)
        backend.stop_on_eos = not fixed_budget
        lat, toks = [], []
        for i in range(n_turns):
            assistant.reset()
            t0 = time.perf_counter()
            assistant.chat(f"Find the python files about topic {i} and summarize.")
            lat.append(time.perf_counter() - t0)
            toks.append(engine.last_metrics.get("new_tokens", 0))
        lat.sort()
        m = assistant.turn_metrics[-1]
        print(json.dumps({
            "metric": "p50 tool-turn latency (prefill + decode + GlobTool + "
                      "continuation)" + (" [fixed decode budget]"
                                         if fixed_budget else " [stop on EOS]"),
            "unit": "ms",
            "p50": round(lat[len(lat) // 2] * 1000, 1),
            "p95": round(lat[int(len(lat) * 0.95)] * 1000, 1),
            "turns": n_turns, "model": model,
            "decode_budget_per_round": max_new,
            "mean_new_tokens_final_round": round(sum(toks) / len(toks), 1),
            "tool_ms_last_turn": round(m["tools"][0]["latency_s"] * 1000, 2)
            if m["tools"] else None,
            "device": "cuda" if has_gpu else "cpu", "data": "synthetic",
        }))


if __name__ == "__main__":
    main()
