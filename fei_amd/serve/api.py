"""OpenAI-style HTTP serving front-end for the local engine.

POST /v1/completions        prompt, max_tokens, temperature, top_k, top_p,
                            stop, n, stream (SSE), echo+logprobs (eval),
                            speculative, cache_prefix
POST /v1/chat/completions   messages=[{role, content}], ... (+ stream SSE)
POST /v1/sessions           open a paged continuous-batching session
POST /v1/sessions/step      advance all sessions (?n=K)
GET  /v1/sessions/{id}      poll result     DELETE /v1/sessions/{id}  close
GET  /v1/models             available ModelSpecs
GET  /health                engine + device status

One engine instance serves requests serially under a lock (the decode hot
path owns the GPU; concurrent multi-session serving is the
PagedSessionManager API, engine/sessions.py — an HTTP multiplexer over it
is a natural extension). The reference has no serving surface at all (its
"server" is the memdir HTTP API); this is engine-native.
"""

from __future__ import annotations

import threading
import time
from typing import List, Optional

from pydantic import BaseModel

from fei_amd.engine.config import MODEL_SPECS
from fei_amd.engine.engine import LocalEngine


class CompletionRequest(BaseModel):
    prompt: str
    model: Optional[str] = None          # informational; engine is fixed
    max_tokens: int = 256
    temperature: float = 0.0
    top_k: int = 0
    top_p: float = 1.0
    stop: Optional[List[str]] = None
    n: int = 1
    stop_on_eos: bool = True
    speculative: Optional[bool] = None
    stream: bool = False
    echo: bool = False
    logprobs: Optional[int] = None
    cache_prefix: bool = True


class ChatMessage(BaseModel):
    role: str
    content: str


class SessionOpen(BaseModel):
    prompt: str
    max_tokens: int = 256


class ChatRequest(BaseModel):
    messages: List[ChatMessage]
    model: Optional[str] = None
    max_tokens: int = 256
    temperature: float = 0.0
    top_k: int = 0
    top_p: float = 1.0
    stream: bool = False


def flatten_chat(messages: List[ChatMessage]) -> str:
    """Minimal chat template for the byte tokenizer (a real checkpoint
    would bring its own template)."""
    parts = [f"{m.role}: {m.content}" for m in messages]
    parts.append("assistant:")
    return "\n".join(parts)


def create_app(engine: Optional[LocalEngine] = None,
               model: str = "llama3-tiny",
               session_blocks: Optional[int] = None, **engine_kwargs):
    from fastapi import FastAPI

    app = FastAPI(title="fei_amd", version="0.1.0")
    eng = engine or LocalEngine.create(model, **engine_kwargs)
    lock = threading.Lock()
    model_name = eng.spec.name

    cache = {"ids": []}     # cross-request prefix cache (LCP, like the
                            # agent backend): repeated/extended prompts
                            # re-prefill only the delta

    def _generate(prompt: str, req) -> dict:
        with lock:
            if req.temperature > 0 and (req.top_k or req.top_p < 1.0):
                out = eng.generate_sampled(
                    prompt, max_new_tokens=req.max_tokens,
                    temperature=req.temperature, top_k=req.top_k,
                    top_p=req.top_p)
                cache["ids"] = []
            else:
                ids = eng.tokenizer.encode(prompt)
                n = 0
                if getattr(req, "cache_prefix", True) and eng.B == 1:
                    limit = min(len(cache["ids"]), len(ids) - 1)
                    while n < limit and ids[n] == cache["ids"][n]:
                        n += 1
                out = eng.generate(
                    ids, max_new_tokens=req.max_tokens,
                    temperature=req.temperature,
                    stop_on_eos=getattr(req, "stop_on_eos", True),
                    speculative=getattr(req, "speculative", None),
                    stop=getattr(req, "stop", None),
                    from_pos=n)
                # drop the final generated token: only sampled, its KV row
                # is never written (same rule as the agent backend)
                cache["ids"] = ids + list(out["token_ids"])[:-1]
        return out

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name,
                "device": str(eng.device),
                "gpu": eng.is_gpu, "hip_graph": eng.use_graph}

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": name, "object": "model",
                          "owned_by": "fei_amd",
                          "active": name == model_name}
                         for name in sorted(MODEL_SPECS)]}

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        t0 = time.time()
        if req.echo and req.max_tokens == 0:
            # evaluation mode (lm-eval pattern): teacher-forced logprobs of
            # the prompt itself, no generation
            ids = eng.tokenizer.encode(req.prompt)
            with lock:
                # loglikelihood overwrites the KV caches from position 0:
                # the cross-request prefix cache no longer describes them
                cache["ids"] = []
                ll = eng.loglikelihood(ids[:1], ids[1:])
            return {
                "id": f"cmpl-{int(t0 * 1000)}",
                "object": "text_completion",
                "model": model_name,
                "choices": [{"index": 0, "text": req.prompt,
                             "finish_reason": "length",
                             "logprobs": {
                                 "token_logprobs": [None] + ll["token_logprobs"],
                                 "is_greedy": ll["is_greedy"],
                             }}],
                "usage": {"prompt_tokens": len(ids),
                          "completion_tokens": 0,
                          "total_tokens": len(ids)},
            }
        if req.stream:
            import json

            from fastapi.responses import StreamingResponse

            def sse():
                with lock:
                    # streaming prefills from position 0: invalidate the
                    # cross-request prefix cache
                    cache["ids"] = []
                    prev = ""
                    for c in eng.generate_stream(
                            req.prompt, max_new_tokens=req.max_tokens,
                            temperature=req.temperature,
                            stop_on_eos=req.stop_on_eos,
                            stop=req.stop):
                        delta = c["text"][len(prev):]
                        prev = c["text"]
                        payload = {"object": "text_completion.chunk",
                                   "model": model_name,
                                   "choices": [{"index": 0, "text": delta,
                                                "finish_reason":
                                                "stop" if c["done"] else None}]}
                        yield f"data: {json.dumps(payload)}\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(sse(), media_type="text/event-stream")
        choices = []
        n_req = max(1, min(int(getattr(req, "n", 1)), 8))
        seed0 = eng.seed
        try:
            for i in range(n_req):
                if n_req > 1 and req.temperature > 0:
                    eng.seed = seed0 + i   # distinct samples per choice
                out = _generate(req.prompt, req)
                choices.append({"index": i, "text": out["text"],
                                "finish_reason": out.get("finish_reason",
                                                         "stop")})
        finally:
            eng.seed = seed0
        n_new = len(out["token_ids"])
        return {
            "id": f"cmpl-{int(t0 * 1000)}",
            "object": "text_completion",
            "created": int(t0),
            "model": model_name,
            "choices": choices,
            "usage": {"prompt_tokens": out.get("prompt_tokens", 0),
                      "completion_tokens": n_new,
                      "total_tokens": out.get("prompt_tokens", 0) + n_new},
            "metrics": {k: v for k, v in eng.last_metrics.items()
                        if isinstance(v, (int, float))},
        }

    @app.post("/v1/chat/completions")
    def chat(req: ChatRequest):
        t0 = time.time()
        if req.stream:
            import json

            from fastapi.responses import StreamingResponse

            prompt = flatten_chat(req.messages)

            def sse():
                with lock:
                    cache["ids"] = []       # stream prefills from position 0
                    prev = ""
                    for c in eng.generate_stream(
                            prompt, max_new_tokens=req.max_tokens,
                            temperature=req.temperature):
                        delta = c["text"][len(prev):]
                        prev = c["text"]
                        payload = {"object": "chat.completion.chunk",
                                   "model": model_name,
                                   "choices": [{"index": 0,
                                                "delta": {"content": delta},
                                                "finish_reason":
                                                "stop" if c["done"] else None}]}
                        yield f"data: {json.dumps(payload)}\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(sse(), media_type="text/event-stream")
        out = _generate(flatten_chat(req.messages), req)
        return {
            "id": f"chatcmpl-{int(t0 * 1000)}",
            "object": "chat.completion",
            "created": int(t0),
            "model": model_name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant",
                                     "content": out["text"]},
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": out.get("prompt_tokens", 0),
                      "completion_tokens": len(out["token_ids"]),
                      "total_tokens": (out.get("prompt_tokens", 0) +
                                       len(out["token_ids"]))},
        }

    # -- multi-session serving (paged continuous batching) -------------------

    state = {"mgr": None}

    def _mgr():
        if state["mgr"] is None:
            from fei_amd.engine.sessions import PagedSessionManager
            state["mgr"] = PagedSessionManager(eng, num_blocks=session_blocks)
        return state["mgr"]

    @app.post("/v1/sessions")
    def session_open(req: SessionOpen):
        from fastapi import HTTPException
        with lock:
            try:
                sid = _mgr().open(req.prompt, max_new_tokens=req.max_tokens)
            except MemoryError as e:
                raise HTTPException(status_code=503, detail=str(e))
        return {"session_id": sid}

    @app.post("/v1/sessions/step")
    def session_step(n: int = 1):
        with lock:
            mgr = _mgr()
            # chunked stepping: ONE host sync for the whole request
            # (engine/sessions.step_chunk) instead of one per step
            advanced = mgr.step_chunk(max(1, n))
        return {"active": advanced,
                "open": len(mgr.sessions),
                "free_blocks": mgr.pool.free_blocks()}

    @app.get("/v1/sessions/{sid}")
    def session_get(sid: int):
        from fastapi import HTTPException
        mgr = _mgr()
        if sid not in mgr.sessions:
            raise HTTPException(status_code=404, detail="no such session")
        return {"session_id": sid, **mgr.result(sid)}

    @app.delete("/v1/sessions/{sid}")
    def session_close(sid: int):
        with lock:
            _mgr().close(sid)
        return {"closed": sid}

    app.state.engine = eng
    return app


def main(argv=None) -> int:
    import argparse

    import uvicorn

    p = argparse.ArgumentParser("fei-api",
                                description="OpenAI-style local serving API")
    p.add_argument("--model", default=None,
                   help="default: llama3-8b on GPU, llama3-tiny on CPU")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8123)
    p.add_argument("--check", action="store_true",
                   help="build the app (loads the model) and exit")
    args = p.parse_args(argv)
    if args.model is None:
        import torch
        args.model = ("llama3-8b" if torch.cuda.is_available()
                      else "llama3-tiny")
    app = create_app(model=args.model)
    if args.check:
        print(f"[fei-api] ok: {app.state.engine.spec.name} on "
              f"{app.state.engine.device}")
        return 0
    uvicorn.run(app, host=args.host, port=args.port)
    return 0


if __name__ == "__main__":
    main()
