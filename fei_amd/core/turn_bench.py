"""Deterministic agent tool-turn latency harness.

Measures the FULL agent turn through the real stack — Assistant.chat ->
prompt render -> tokenize -> prefill -> hipGraph decode -> tool-call
parse -> GlobTool execution -> continuation prefill+decode — with the
turn STRUCTURE made deterministic by injecting the tool round (weights
are random-init, so the model cannot emit valid JSON itself; the
scripted-turn harness SURVEY.md §7 hard-part 3 calls for). ALL latency
comes from real engine + tool work.

This is the second half of the BASELINE metric (p50 tool-turn latency);
bench.py surfaces the result in its JSON line so the driver records it
(VERDICT r01 missing #6).
"""

from __future__ import annotations

import os
import time
from typing import Dict, List

from fei_amd.core.backends import LocalBackend


class TurnBackend(LocalBackend):
    """Real engine inference; deterministic tool-call injection on the
    first round of each turn."""

    def __init__(self, engine, max_new: int, tool_name: str = "GlobTool",
                 tool_input=None):
        super().__init__(engine=engine, stop_on_eos=False)
        self.max_new = max_new
        self.tool_name = tool_name
        # BOUNDED tool output: a repo-wide '**/*.py' glob returns hundreds
        # of paths, overflowing the engine context so the continuation
        # round's decode budget collapses to 1 token — the turn latency
        # then measures a degenerate turn. A package-dir glob (~10 files)
        # keeps both rounds decoding their full budget.
        if tool_input is None:
            import fei_amd.core as _core
            tool_input = {"pattern": "*.py",
                          "path": os.path.dirname(_core.__file__)}
        self.tool_input = tool_input
        self._round = 0

    def complete(self, messages, tools=None, system=None, max_tokens=4000,
                 temperature=0.0):
        out = super().complete(messages, tools, system,
                               max_tokens=self.max_new,
                               temperature=temperature)
        self._round += 1
        if self._round % 2 == 1:        # first round of a turn: call a tool
            out.tool_calls = [{"id": f"c{self._round}",
                               "name": self.tool_name,
                               "input": dict(self.tool_input)}]
        else:
            out.tool_calls = []
        return out


def measure_tool_turns(engine, n_turns: int = 9, max_new: int = 96,
                       ) -> Dict[str, object]:
    """Run ``n_turns`` full tool-turns on ``engine`` and return latency
    percentiles. Each turn is: user msg -> round 1 (prefill+decode, tool
    call injected) -> GlobTool -> round 2 (continuation prefill+decode).
    The decode budget is FIXED (stop_on_eos off) so random-init weights
    cannot shorten a turn by sampling EOS early."""
    from fei_amd.core.assistant import Assistant
    from fei_amd.tools.code import create_code_tools
    from fei_amd.tools.registry import ToolRegistry

    registry = ToolRegistry()
    create_code_tools(registry)
    backend = TurnBackend(engine, max_new)
    assistant = Assistant(provider="local", tool_registry=registry,
                          backend=backend)
    lat: List[float] = []
    # one untimed warmup turn (first GlobTool call populates its dir cache)
    assistant.reset()
    assistant.chat("Find the python files about warmup and summarize.")
    for i in range(n_turns):
        assistant.reset()
        t0 = time.perf_counter()
        assistant.chat(f"Find the python files about topic {i} and summarize.")
        lat.append(time.perf_counter() - t0)
    lat.sort()
    return {
        "tool_turn_p50_s": round(lat[len(lat) // 2], 4),
        "tool_turn_p95_s": round(lat[min(int(len(lat) * 0.95),
                                         len(lat) - 1)], 4),
        "tool_turn_n": n_turns,
        "tool_turn_decode_budget": max_new,
        "tokenizer": type(engine.tokenizer).__name__,
    }
