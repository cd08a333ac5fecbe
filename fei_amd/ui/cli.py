"""fei CLI: chat loop, single message, continuous task mode, subcommands.

Parity: reference fei/ui/cli.py (786 LoC): chat history JSON in
``~/.fei/history.json`` capped at 100 entries (:68-137), chat_loop with
exit/quit/clear/history commands (:176-269), empty-response recovery by
scraping tool output (:240-264), single-message mode (:285-334), continuous
task mode injecting the [TASK_COMPLETE] convention (:336-361), argparse
subcommands ask/history/mcp/search (:386-441).

Reference defect NOT replicated: the hardcoded fallback Brave API key
(cli.py:589) — ``fei ask`` degrades to a plain local-model answer when no
search backend is configured.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
from typing import Any, Dict, List, Optional

from fei_amd.utils.config import get_config
from fei_amd.utils.logging import get_logger, setup_logging

logger = get_logger("ui.cli")

HISTORY_PATH = os.path.join(os.path.expanduser("~"), ".fei", "history.json")
HISTORY_CAP = 100


class ChatHistory:
    def __init__(self, path: Optional[str] = None):
        # resolved at call time so tests (and HOME changes) take effect
        self.path = path or HISTORY_PATH
        self.entries: List[Dict[str, Any]] = []
        self._load()

    def _load(self) -> None:
        try:
            with open(self.path, "r", encoding="utf-8") as f:
                self.entries = json.load(f)[-HISTORY_CAP:]
        except (OSError, json.JSONDecodeError):
            self.entries = []

    def add(self, prompt: str, response: str) -> None:
        self.entries.append({"ts": time.time(), "prompt": prompt,
                             "response": response[:4000]})
        self.entries = self.entries[-HISTORY_CAP:]
        try:
            os.makedirs(os.path.dirname(self.path), exist_ok=True)
            with open(self.path, "w", encoding="utf-8") as f:
                json.dump(self.entries, f)
        except OSError:
            pass

    def clear(self) -> None:
        self.entries = []
        try:
            os.unlink(self.path)
        except OSError:
            pass


class CLI:
    def __init__(self, provider: Optional[str] = None,
                 model: Optional[str] = None,
                 with_memory: bool = True,
                 script: Optional[List[Dict[str, Any]]] = None,
                 api_key: Optional[str] = None):
        from fei_amd.core.assistant import Assistant
        from fei_amd.tools.code import create_code_tools
        from fei_amd.tools.registry import ToolRegistry

        self.config = get_config()
        registry = ToolRegistry()
        create_code_tools(registry)
        if with_memory:
            from fei_amd.tools.memory_tools import create_memory_tools
            create_memory_tools(registry)
        kwargs: Dict[str, Any] = {}
        if script is not None:
            kwargs["script"] = script
            provider = "scripted"
        self.assistant = Assistant(config=self.config, provider=provider,
                                   model=model, api_key=api_key,
                                   tool_registry=registry, **kwargs)
        self.history = ChatHistory()

    # -- chat loop -----------------------------------------------------------

    def chat_loop(self, stdin=None, stdout=None) -> int:
        stdin = stdin or sys.stdin
        stdout = stdout or sys.stdout
        print("fei (MI355X-local). Commands: exit, quit, clear, history.",
              file=stdout)
        while True:
            try:
                stdout.write("you> ")
                stdout.flush()
                line = stdin.readline()
            except KeyboardInterrupt:
                print("", file=stdout)
                return 0
            if not line:
                return 0
            message = line.strip()
            if not message:
                continue
            if message in ("exit", "quit"):
                return 0
            if message == "clear":
                self.assistant.reset()
                print("(conversation cleared)", file=stdout)
                continue
            if message == "history":
                for e in self.history.entries[-10:]:
                    print(f"  > {e['prompt'][:70]}", file=stdout)
                continue
            response = self.assistant.ask(message)
            if not response:
                response = self.assistant.conversation.scrape_tool_output() \
                    or "(no response)"
            print(f"fei> {response}", file=stdout)
            self.history.add(message, response)

    def single_message(self, message: str, task: bool = False,
                       max_iterations: int = 10) -> str:
        if task:
            from fei_amd.core.task_executor import TaskExecutor
            result = TaskExecutor(self.assistant).execute_task(
                message, max_iterations=max_iterations)
            response = result["final_response"]
        else:
            response = self.assistant.ask(message)
        self.history.add(message, response)
        return response


def handle_ask_command(question: str, provider: Optional[str],
                       model: Optional[str]) -> str:
    """Search-augmented one-shot (reference: cli.py:623-728). With no search
    backend (offline), answers from the local model alone."""
    context = ""
    try:
        from fei_amd.core.mcp import MCPManager
        mgr = MCPManager()
        results = mgr.brave_search.search(question, count=5)
        if isinstance(results, list) and results:
            context = "\n".join(
                f"- {r.get('title')}: {r.get('description')} ({r.get('url')})"
                for r in results[:5])
    except Exception:  # noqa: BLE001 — offline / no key: degrade gracefully
        context = ""
    cli = CLI(provider=provider, model=model, with_memory=False)
    prompt = question if not context else (
        f"Context from web search:\n{context}\n\nQuestion: {question}")
    return cli.single_message(prompt)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="fei",
                                description="MI355X-local coding assistant")
    from fei_amd import __version__
    p.add_argument("--version", action="version",
                   version=f"fei_amd {__version__}")
    p.add_argument("--provider", choices=["local", "stub", "scripted"],
                   default=None)
    p.add_argument("--model", default=None)
    p.add_argument("--api-key", default=None,
                   help="API key (reference-compatible; unused by local)")
    p.add_argument("--debug", action="store_true", help="debug logging")
    p.add_argument("--message", "-m", help="single message, print reply, exit")
    p.add_argument("--task", help="run a continuous task to completion")
    p.add_argument("--max-iterations", type=int, default=10)
    p.add_argument("--textual", action="store_true", help="launch the TUI")
    p.add_argument("--no-memory", action="store_true")
    p.add_argument("--stats", action="store_true",
                   help="print per-turn metrics (LLM/tool latencies) as JSON")
    sub = p.add_subparsers(dest="cmd")

    a = sub.add_parser("ask", help="search-augmented one-shot question")
    a.add_argument("question", nargs="+")

    h = sub.add_parser("history", help="show recent chat history")
    h.add_argument("--limit", type=int, default=20)
    h.add_argument("--clear", action="store_true", help="clear the history")
    h.add_argument("--load", type=int, default=None,
                   help="print one entry (prompt + reply) by index")

    m = sub.add_parser("mcp", help="list configured MCP servers")

    s = sub.add_parser("search", help="search memdir memories")
    s.add_argument("query", nargs="+")

    api = sub.add_parser("api", help="OpenAI-style local serving API")
    api.add_argument("--model", default="llama3-8b")
    api.add_argument("--host", default="127.0.0.1")
    api.add_argument("--api-port", type=int, default=8123)
    b = sub.add_parser("serve", help="run the memdir HTTP server")
    b.add_argument("--port", type=int, default=5000)

    sub.add_parser("doctor", help="environment and kernel diagnostics")
    return p


def run_doctor() -> int:
    """Environment diagnostics: torch/ROCm, GPU, kernel library, hipcc,
    memdir base, config sources."""
    import shutil
    import subprocess

    import torch

    from fei_amd import __version__, ops
    from fei_amd.memdir import utils as mu

    rows = [("fei_amd", __version__),
            ("python", sys.version.split()[0]),
            ("torch", torch.__version__),
            ("gpu", "yes" if torch.cuda.is_available() else "no")]
    if torch.cuda.is_available():
        rows.append(("device", torch.cuda.get_device_name(0)))
        free, total = torch.cuda.mem_get_info(0)
        rows.append(("hbm", f"{free / 2**30:.0f} / {total / 2**30:.0f} GiB free"))
    rows.append(("kernel lib", ops._LIB_PATH if ops.kernels_available()
                 else f"MISSING ({ops._LIB_PATH}) — run python -m fei_amd.ops.build"))
    hipcc = shutil.which("hipcc")
    if hipcc:
        try:
            ver = subprocess.run([hipcc, "--version"], capture_output=True,
                                 text=True, timeout=20).stdout.splitlines()[0]
        except (subprocess.SubprocessError, IndexError, OSError):
            ver = hipcc
        rows.append(("hipcc", ver))
    else:
        rows.append(("hipcc", "not found (kernel rebuilds unavailable)"))
    rows.append(("memdir base", mu.get_memdir_base()))
    for env in ("FEI_WEIGHTS", "FEI_TOKENIZER"):
        val = os.environ.get(env)
        if val:
            exists = os.path.exists(val)
            rows.append((env.lower(), val if exists
                         else f"MISSING ({val})"))
    try:
        import fastapi  # noqa: F401
        import uvicorn  # noqa: F401
        rows.append(("serving api", "available (fei api)"))
    except ImportError as e:
        rows.append(("serving api", f"unavailable ({e.name} missing)"))
    try:
        import sentencepiece  # noqa: F401
        rows.append(("tokenizers", "byte + sentencepiece"))
    except ImportError:
        rows.append(("tokenizers", "byte only"))
    cfg = get_config()
    rows.append(("provider", cfg.get("llm.provider")))
    rows.append(("model", cfg.get("llm.model")))
    ok = True
    for key, value in rows:
        print(f"{key:12s} {value}")
        if isinstance(value, str) and value.startswith("MISSING"):
            ok = False
    if torch.cuda.is_available() and not ops.kernels_available():
        print("ERROR: GPU present but kernel library missing — GPU execution "
              "will refuse to run (no eager fallback).")
        ok = False
    return 0 if ok else 1


def main(argv: Optional[List[str]] = None) -> int:
    setup_logging()
    parser = build_parser()
    args = parser.parse_args(argv)
    if getattr(args, "debug", False):
        import logging
        logging.getLogger("fei_amd").setLevel(logging.DEBUG)

    if args.cmd == "ask":
        print(handle_ask_command(" ".join(args.question), args.provider,
                                 args.model))
        return 0
    if args.cmd == "history":
        hist = ChatHistory()
        if args.clear:
            hist.clear()
            print("history cleared")
            return 0
        if args.load is not None:
            try:
                e = hist.entries[args.load]
            except IndexError:
                print(f"no history entry {args.load}")
                return 1
            print(f"> {e['prompt']}\n{e['response']}")
            return 0
        for i, e in enumerate(hist.entries[-args.limit:]):
            stamp = time.strftime("%m-%d %H:%M", time.localtime(e["ts"]))
            print(f"[{i}] [{stamp}] {e['prompt'][:70]}")
        return 0
    if args.cmd == "mcp":
        from fei_amd.core.mcp import MCPClient
        for name in MCPClient().list_servers():
            print(name)
        return 0
    if args.cmd == "search":
        from fei_amd.memdir.search import format_results, search
        print(format_results(search(" ".join(args.query)), "compact"))
        return 0
    if args.cmd == "api":
        from fei_amd.serve.api import main as api_main
        return api_main(["--model", args.model, "--host", args.host,
                         "--port", str(args.api_port)])
    if args.cmd == "serve":
        from fei_amd.memdir.run_server import main as serve_main
        return serve_main(["--port", str(args.port)])
    if args.cmd == "doctor":
        return run_doctor()

    if args.textual:
        from fei_amd.ui.tui import main as tui_main
        app = tui_main(provider=args.provider, model=args.model)
        app.run()
        return 0

    cli = CLI(provider=args.provider, model=args.model,
              with_memory=not args.no_memory, api_key=args.api_key)
    if args.task:
        print(cli.single_message(args.task, task=True,
                                 max_iterations=args.max_iterations))
        return 0
    if args.message:
        print(cli.single_message(args.message))
        if args.stats:
            print(json.dumps(cli.assistant.turn_metrics, default=str))
        return 0
    return cli.chat_loop()


if __name__ == "__main__":
    raise SystemExit(main())
