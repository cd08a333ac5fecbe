"""LLM backends for the Assistant.

Where the reference calls remote APIs through LiteLLM
(fei/core/assistant.py:527-530), fei_amd resolves a *backend*:

  - ``StubBackend``     — deterministic echo model for CPU plumbing tests
                          (the reference's mocked-litellm test pattern,
                          fei/tests/test_litellm.py, promoted to a
                          first-class backend; BASELINE.json configs[0])
  - ``ScriptedBackend`` — plays a pre-written sequence of turns, including
                          tool calls: makes agent latency/throughput
                          measurable with random-init weights
  - ``LocalBackend``    — the MI355X inference engine (fei_amd.engine):
                          HIP kernels, hipGraph decode, TP over RCCL

Backend protocol: ``complete(messages, tools, system, max_tokens,
temperature) -> BackendResponse``. ``tool_calls`` entries are
``{"id": str, "name": str, "input": dict}`` (Anthropic shape — the
reference's conversation format we keep, assistant.py:266-303).
"""

from __future__ import annotations

import json
import re
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from fei_amd.utils.logging import get_logger

logger = get_logger("core.backends")


@dataclass
class BackendResponse:
    content: str
    tool_calls: List[Dict[str, Any]] = field(default_factory=list)
    usage: Dict[str, int] = field(default_factory=dict)
    raw: Any = None


class Backend:
    name = "base"

    def complete(
        self,
        messages: List[Dict[str, Any]],
        tools: Optional[List[Dict[str, Any]]] = None,
        system: Optional[str] = None,
        max_tokens: int = 4000,
        temperature: float = 0.0,
    ) -> BackendResponse:
        raise NotImplementedError

    def close(self) -> None:
        pass


class StubBackend(Backend):
    """Echo backend: replies deterministically, never calls tools unless the
    user message contains an explicit ``CALL_TOOL`` directive of the form
    ``CALL_TOOL <name> <json-args>`` (useful in tests)."""

    name = "stub"

    def complete(self, messages, tools=None, system=None, max_tokens=4000,
                 temperature=0.0) -> BackendResponse:
        last_user = ""
        for msg in reversed(messages):
            if msg.get("role") == "user":
                content = msg.get("content", "")
                if isinstance(content, list):  # tool-result blocks
                    parts = [b.get("content", "") for b in content
                             if isinstance(b, dict) and b.get("type") == "tool_result"]
                    last_user = "\n".join(str(p) for p in parts)
                else:
                    last_user = str(content)
                break
        m = re.match(r"\s*CALL_TOOL\s+(\S+)\s+(\{.*\})\s*$", last_user, re.S)
        if m and tools is not None:
            try:
                args = json.loads(m.group(2))
                return BackendResponse(
                    content="",
                    tool_calls=[{"id": f"call_{uuid.uuid4().hex[:8]}",
                                 "name": m.group(1), "input": args}],
                    usage={"input_tokens": len(last_user.split()),
                           "output_tokens": 8},
                )
            except json.JSONDecodeError:
                pass
        reply = f"[stub] You said: {last_user[:2000]}"
        return BackendResponse(
            content=reply,
            usage={"input_tokens": len(last_user.split()), "output_tokens": len(reply.split())},
        )


class ScriptedBackend(Backend):
    """Plays a scripted list of BackendResponse-like dicts in order; repeats
    the last one when exhausted. Deterministic agent turns for benchmarks
    (SURVEY.md §7 hard-part 3)."""

    name = "scripted"

    def __init__(self, script: List[Dict[str, Any]]):
        self.script = list(script)
        self._i = 0

    def complete(self, messages, tools=None, system=None, max_tokens=4000,
                 temperature=0.0) -> BackendResponse:
        if not self.script:
            return BackendResponse(content="")
        step = self.script[min(self._i, len(self.script) - 1)]
        self._i += 1
        calls = []
        for tc in step.get("tool_calls", []):
            calls.append({"id": tc.get("id") or f"call_{uuid.uuid4().hex[:8]}",
                          "name": tc["name"], "input": tc.get("input", {})})
        return BackendResponse(content=step.get("content", ""), tool_calls=calls,
                               usage=step.get("usage", {}))


# Tool-call wire format for the local model: the model is prompted to emit
#   <tool_call>{"name": ..., "arguments": {...}}</tool_call>
# blocks; we parse them out of the generated text.
_TOOL_CALL_RE = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.S)


def extract_tool_call_blocks(text: str) -> List[Dict[str, Any]]:
    calls: List[Dict[str, Any]] = []
    for m in _TOOL_CALL_RE.finditer(text):
        try:
            obj = json.loads(m.group(1))
        except json.JSONDecodeError:
            continue
        name = obj.get("name")
        if not name:
            continue
        calls.append({
            "id": f"call_{uuid.uuid4().hex[:8]}",
            "name": name,
            "input": obj.get("arguments", obj.get("input", {})) or {},
        })
    return calls


def strip_tool_call_blocks(text: str) -> str:
    return _TOOL_CALL_RE.sub("", text).strip()


class LocalBackend(Backend):
    """The MI355X-native engine backend.

    Prompts are rendered with a simple chat template; tool schemas are
    embedded in the system prompt and tool calls parsed from
    ``<tool_call>`` blocks. The engine does prefill with the HIP flash
    kernel and decode with the hipGraph-captured step
    (fei_amd.engine.engine.LocalEngine).
    """

    name = "local"

    def __init__(self, engine=None, model: str = "llama3-8b",
                 stop_on_eos: bool = True, **engine_kwargs):
        if engine is None:
            import torch

            from fei_amd.engine.config import get_spec
            from fei_amd.engine.engine import LocalEngine
            spec = get_spec(model)
            if not torch.cuda.is_available() and spec.hidden_size >= 2048:
                raise RuntimeError(
                    f"provider 'local' with {model} needs a GPU (a CPU run "
                    "would materialise the full model in host RAM at fp32). "
                    "Use --provider stub for plumbing, or --model "
                    "llama3-tiny for CPU experiments.")
            engine = LocalEngine.create(model, **engine_kwargs)
        self.engine = engine
        self.stop_on_eos = stop_on_eos
        # prefix cache: token ids currently materialised in the KV caches
        self._cached_ids: List[int] = []

    @staticmethod
    def render_prompt(messages, tools=None, system=None) -> str:
        parts: List[str] = []
        sys_text = system or "You are a helpful coding assistant."
        if tools:
            # compact tool list: name, one-line description, required args.
            # (The byte-level tokenizer costs ~1 token/char; full JSON
            # schemas would eat the context. Argument validation happens in
            # the registry regardless.)
            lines = []
            for t in tools:
                req = ",".join(t.get("input_schema", {}).get("required", []))
                desc = t.get("description", "").split(". ")[0][:100]
                lines.append(f"- {t['name']}({req}): {desc}")
            sys_text += (
                "\nYou can call tools by emitting "
                '<tool_call>{"name": "...", "arguments": {...}}</tool_call>.'
                "\nAvailable tools:\n" + "\n".join(lines)
            )
        parts.append(f"<|system|>\n{sys_text}\n")
        for msg in messages:
            role = msg.get("role", "user")
            content = msg.get("content", "")
            if isinstance(content, list):
                chunks = []
                for block in content:
                    if isinstance(block, dict):
                        if block.get("type") == "tool_result":
                            chunks.append(f"<tool_result>{block.get('content','')}</tool_result>")
                        elif block.get("type") == "text":
                            chunks.append(str(block.get("text", "")))
                        elif block.get("type") == "tool_use":
                            chunks.append(
                                "<tool_call>" + json.dumps(
                                    {"name": block.get("name"),
                                     "arguments": block.get("input", {})}) + "</tool_call>")
                content = "\n".join(chunks)
            parts.append(f"<|{role}|>\n{content}\n")
        parts.append("<|assistant|>\n")
        return "".join(parts)

    def complete(self, messages, tools=None, system=None, max_tokens=4000,
                 temperature=0.0) -> BackendResponse:
        prompt = self.render_prompt(messages, tools, system)
        ids = self.engine.tokenizer.encode(prompt)
        # prefix cache: a turn's continuation prompt extends the previous
        # one, so only the delta needs prefilling (the engine's KV caches
        # already hold the prefix + its generated tokens are NOT part of
        # the rendered prompt — cache only the prompt prefix).
        # longest common prefix (not all-or-nothing: generated specials
        # may not re-render byte-identically, but everything before them
        # still reuses the cache)
        n_common = 0
        limit = min(len(self._cached_ids), len(ids) - 1)
        while n_common < limit and ids[n_common] == self._cached_ids[n_common]:
            n_common += 1
        out = self.engine.generate(ids, max_new_tokens=max_tokens,
                                   temperature=temperature,
                                   stop_on_eos=self.stop_on_eos,
                                   from_pos=n_common)
        # extend the cache through the GENERATED tokens too: the response
        # is already in the KV caches, and the next turn's rendered prompt
        # repeats it byte-for-byte (byte tokenizer: decode->render->encode
        # is identity for non-special tokens). If the rendering ever
        # differs (specials got dropped), the next turn's prefix compare
        # simply fails and we re-prefill — correctness never depends on it.
        # The FINAL generated token was only sampled — its KV row is written
        # when it becomes the next decode step's input, which never happens
        # for the last token of a turn. Drop it unconditionally (covers both
        # the eos and the length-finish case): letting the next turn's LCP
        # reach a position with no KV silently corrupts attention
        # (ADVICE r01, medium).
        gen = list(out["token_ids"])[:-1]
        self._cached_ids = ids + gen
        text = out["text"]
        calls = extract_tool_call_blocks(text)
        return BackendResponse(
            content=strip_tool_call_blocks(text),
            tool_calls=calls,
            usage={"input_tokens": out.get("prompt_tokens", 0),
                   "output_tokens": out.get("new_tokens", 0),
                   "cached_prefix": out.get("cached_prefix", 0),
                   "decode_tok_s": round(out.get("decode_tok_s", 0.0), 1),
                   "prefill_tok_s": round(out.get("prefill_tok_s", 0.0), 1)},
            raw=out,
        )

    def close(self) -> None:
        self.engine.shutdown()


def create_backend(provider: str, model: str = "llama3-8b", **kwargs) -> Backend:
    """Provider -> backend resolution (the reference's ProviderManager role,
    assistant.py:25-111, re-aimed at local backends)."""
    provider = (provider or "local").lower()
    if provider == "stub":
        return StubBackend()
    if provider == "scripted":
        return ScriptedBackend(kwargs.get("script", []))
    if provider == "local":
        return LocalBackend(engine=kwargs.get("engine"), model=model,
                            **kwargs.get("engine_kwargs", {}))
    raise ValueError(f"unknown provider {provider!r} (expected stub|scripted|local)")
