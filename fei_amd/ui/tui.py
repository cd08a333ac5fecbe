"""Textual TUI: chat app with message panels and the /mem command suite.

Parity: reference fei/ui/textual_chat.py (1,070 LoC) — message widget
hierarchy and panels (:48-117), memory-command autocomplete suggester
(:119-229), the /mem command set (help/list/search/view/save/tag/server —
:557-970) calling the memory tool handlers directly, background assistant
processing (:1002-1031), and ``main()`` returning the App for the caller
to ``.run()`` (:1044-1062). This build adds a status sidebar (provider,
model, turn metrics) and slash commands beyond /mem.
"""

from __future__ import annotations

import time
from typing import List, Optional

from textual.app import App, ComposeResult
from textual.containers import Horizontal, Vertical, VerticalScroll
from textual.suggester import Suggester
from textual.widgets import Footer, Header, Input, Static

MEM_COMMANDS = [
    "/mem help", "/mem list", "/mem search ", "/mem save ",
    "/mem tag ", "/mem server start", "/mem server stop", "/mem server status",
    "/mem index", "/mem semantic ", "/mem keyword ", "/mem view ",
]

SLASH_COMMANDS = [
    "/help", "/clear", "/history", "/stats", "/quit",
]

MEM_HELP = """\
/mem help                 this help
/mem list [folder]        list memories
/mem search <query>       query-language search
/mem semantic <query>     embedding-index search (GPU)
/mem keyword <query>      bm25 FTS search
/mem index                rebuild the embedding index
/mem view <id>            show one memory
/mem save <subject>       save the conversation as a memory
/mem tag <id> <tags>      set tags on a memory
/mem server start|stop|status"""

APP_HELP = MEM_HELP + """

/help                     this help
/clear                    clear the conversation
/history                  show the turn history
/stats                    per-turn metrics (tokens, latency, tools)
/quit                     exit"""


class CommandSuggester(Suggester):
    """Autocomplete for slash + /mem commands AND memory ids for
    /mem view|tag (reference: textual_chat.py:119-229). Memory ids are
    fetched lazily from the memory tools and cached briefly."""

    def __init__(self, app: Optional["FeiChatApp"] = None):
        super().__init__(use_cache=False, case_sensitive=False)
        self._app = app
        self._ids: List[str] = []
        self._ids_at = 0.0

    def _memory_ids(self) -> List[str]:
        if time.time() - self._ids_at < 5.0:
            return self._ids
        self._ids_at = time.time()
        try:
            tools = self._app._tools() if self._app else None
            out = tools.list({"folder": ""}) if tools else {}
            self._ids = [m["memory_id"] for m in out.get("memories", [])][:50]
        except Exception:
            self._ids = []
        return self._ids

    async def get_suggestion(self, value: str) -> Optional[str]:
        if not value.startswith("/"):
            return None
        # memory-id completion for view/tag
        for stem in ("/mem view ", "/mem tag "):
            if value.startswith(stem):
                frag = value[len(stem):]
                for mid in self._memory_ids():
                    if mid.startswith(frag) and mid != frag:
                        return stem + mid
                return None
        for cmd in MEM_COMMANDS + SLASH_COMMANDS:
            if cmd.startswith(value) and cmd != value:
                return cmd
        return None


MemCommandSuggester = CommandSuggester        # back-compat alias


# -- message widget hierarchy (reference: textual_chat.py:48-117) ------------

class ChatMessage(Static):
    """Base message panel; subclasses set the border title and style."""

    role = "system"
    prefix = ""

    def __init__(self, text: str):
        super().__init__(text)
        self.add_class(self.role)
        self.border_title = self.prefix + time.strftime("%H:%M:%S")


class UserMessage(ChatMessage):
    role = "user"
    prefix = "you · "


class AssistantMessage(ChatMessage):
    role = "assistant"
    prefix = "fei · "


class ToolMessage(ChatMessage):
    role = "tool"
    prefix = "tool · "


class SystemMessage(ChatMessage):
    role = "system"


class StatusPanel(Static):
    """Sidebar: provider/model and rolling turn metrics."""

    def update_from(self, assistant) -> None:
        lines = ["fei_amd", ""]
        if assistant is not None:
            prov = getattr(assistant, "provider", None) or "?"
            model = getattr(assistant, "model", None) or ""
            lines += [f"provider  {prov}", f"model     {model}", ""]
            metrics = getattr(assistant, "turn_metrics", [])
            if metrics:
                m = metrics[-1]
                lines += ["last turn:",
                          f"  {m.get('latency_s', 0):.2f}s",
                          f"  tools: {len(m.get('tools', []))}",
                          f"  in/out: {m.get('input_tokens', 0)}"
                          f"/{m.get('output_tokens', 0)}"]
            lines += ["", f"turns     {len(metrics)}"]
        self.update("\n".join(lines))


class FeiChatApp(App):
    CSS = """
    #body { height: 1fr; }
    #chat { height: 1fr; }
    #status { width: 26; border-left: solid $primary; padding: 0 1;
              display: none; }
    #status.visible { display: block; }
    ChatMessage { padding: 0 1; margin-bottom: 1; border: round $primary 30%;
                  border-title-align: left; }
    .user { background: $boost; border: round $secondary; }
    .assistant { border: round $primary; }
    .tool { color: $text-muted; border: round $primary 30%; }
    .system { color: $text-muted; border: none; }
    #thinking { color: $text-muted; padding: 0 1; display: none; }
    #thinking.active { display: block; }
    """
    BINDINGS = [
        ("ctrl+c", "quit", "Quit"),
        ("ctrl+l", "clear", "Clear"),
        ("f2", "toggle_status", "Status"),
    ]

    def __init__(self, assistant=None, memory_tools=None, **kwargs):
        super().__init__(**kwargs)
        self.assistant = assistant
        self.memory_tools = memory_tools

    def _tools(self):
        if self.memory_tools is None:
            from fei_amd.tools.memory_tools import MemoryTools
            self.memory_tools = MemoryTools()
        return self.memory_tools

    # -- layout --------------------------------------------------------------

    def compose(self) -> ComposeResult:
        yield Header(show_clock=True)
        with Horizontal(id="body"):
            with Vertical():
                yield VerticalScroll(id="chat")
                yield Static("thinking…", id="thinking")
            yield StatusPanel(id="status")
        yield Input(placeholder="Message (or /help, /mem ...)",
                    suggester=CommandSuggester(self), id="input")
        yield Footer()

    def add_message(self, text: str, kind: str = "assistant") -> None:
        chat = self.query_one("#chat", VerticalScroll)
        cls = {"user": UserMessage, "assistant": AssistantMessage,
               "tool": ToolMessage, "system": SystemMessage}[kind]
        chat.mount(cls(text))
        chat.scroll_end(animate=False)

    def action_clear(self) -> None:
        if self.assistant is not None:
            self.assistant.reset()
        self.query_one("#chat", VerticalScroll).remove_children()

    def action_toggle_status(self) -> None:
        panel = self.query_one("#status", StatusPanel)
        panel.toggle_class("visible")
        panel.update_from(self.assistant)

    def _set_thinking(self, on: bool) -> None:
        try:
            self.query_one("#thinking", Static).set_class(on, "active")
        except Exception:
            pass

    # -- input handling ------------------------------------------------------

    async def on_input_submitted(self, event: Input.Submitted) -> None:
        text = event.value.strip()
        event.input.value = ""
        if not text:
            return
        self.add_message(text, "user")
        if text.startswith("/mem"):
            self.add_message(self.handle_memory_command(text), "system")
            return
        if text.startswith("/"):
            out = self.handle_slash_command(text)
            if out is not None:
                self.add_message(out, "system")
            return
        self.run_worker(self.process_with_assistant(text), exclusive=True)

    def handle_slash_command(self, text: str) -> Optional[str]:
        cmd = text.split()[0]
        if cmd == "/help":
            return APP_HELP
        if cmd == "/clear":
            self.action_clear()
            return "(cleared)"
        if cmd == "/quit":
            self.exit()
            return None
        if cmd == "/history":
            if self.assistant is None:
                return "(no assistant)"
            msgs = self.assistant.conversation.messages
            lines = []
            for m in msgs[-20:]:
                content = m.get("content", "")
                if isinstance(content, list):
                    content = " ".join(
                        str(b.get("text", b.get("content", "")))[:40]
                        for b in content if isinstance(b, dict))
                lines.append(f"{m.get('role', '?'):<10} {str(content)[:70]}")
            return "\n".join(lines) or "(empty)"
        if cmd == "/stats":
            if self.assistant is None:
                return "(no assistant)"
            lines = []
            for i, m in enumerate(self.assistant.turn_metrics[-10:]):
                tools = ",".join(t.get("name", "?")
                                 for t in m.get("tools", []))
                lines.append(f"turn {i}: {m.get('latency_s', 0):.2f}s "
                             f"in {m.get('input_tokens', 0)} "
                             f"out {m.get('output_tokens', 0)} "
                             f"[{tools}]")
            return "\n".join(lines) or "(no turns yet)"
        return f"unknown command {cmd} (try /help)"

    async def process_with_assistant(self, text: str) -> None:
        """Background turn (reference: textual_chat.py:1002-1031)."""
        if self.assistant is None:
            self.add_message("(no assistant configured)", "system")
            return
        import asyncio
        self._set_thinking(True)
        try:
            loop = asyncio.get_running_loop()
            response = await loop.run_in_executor(None, self.assistant.ask,
                                                  text)
        finally:
            self._set_thinking(False)
        # surface the turn's tool calls like the reference's tool panels
        metrics = getattr(self.assistant, "turn_metrics", [])
        if metrics:
            for t in metrics[-1].get("tools", []):
                self.add_message(
                    f"{t.get('name', '?')}({t.get('latency_s', 0):.2f}s)",
                    "tool")
        self.add_message(response or "(no response)")
        panel = self.query_one("#status", StatusPanel)
        if panel.has_class("visible"):
            panel.update_from(self.assistant)

    # -- /mem commands (reference: textual_chat.py:557-970) -------------------

    def handle_memory_command(self, text: str) -> str:
        tools = self._tools()
        parts = text.split()
        cmd = parts[1] if len(parts) > 1 else "help"
        rest = parts[2:]
        try:
            if cmd == "help":
                return MEM_HELP
            if cmd == "list":
                out = tools.list({"folder": rest[0] if rest else ""})
                lines = [f"{m['memory_id']} [{m['status']}] {m['subject']}"
                         for m in out.get("memories", [])]
                return "\n".join(lines) or "(no memories)"
            if cmd == "search":
                out = tools.search({"query": " ".join(rest),
                                    "with_content": False})
                return "\n".join(
                    f"{m['metadata']['unique']} {m.get('headers', {}).get('Subject', '')}"
                    for m in out.get("results", [])) or "(no matches)"
            if cmd == "keyword":
                out = tools.keyword_search({"query": " ".join(rest),
                                            "with_content": False})
                return "\n".join(
                    f"{m['metadata']['unique']} {m.get('headers', {}).get('Subject', '')}"
                    for m in out.get("results", [])) or "(no matches)"
            if cmd == "semantic":
                out = tools.semantic_search({"query": " ".join(rest)})
                if "error" in out:
                    return out["error"]
                return "\n".join(
                    f"{m.get('score')} {m.get('headers', {}).get('Subject', '')}"
                    for m in out.get("results", [])) or "(no matches)"
            if cmd == "index":
                return f"indexed {tools.index_build({}).get('indexed', 0)} memories"
            if cmd == "view":
                if not rest:
                    return "usage: /mem view <id>"
                mem = tools.view({"memory_id": rest[0]})
                if "error" in mem:
                    return mem["error"]
                hdrs = "\n".join(f"{k}: {v}" for k, v in mem["headers"].items())
                return f"{hdrs}\n---\n{mem['content']}"
            if cmd == "save":
                if self.assistant is None:
                    return "(no conversation)"
                from fei_amd.tools.memory_tools import MemoryManager
                mgr = MemoryManager(base=tools.base)
                out = mgr.save_conversation(self.assistant.conversation.messages,
                                            subject=" ".join(rest) or None)
                return f"saved as {out.get('memory_id')}"
            if cmd == "tag":
                if len(rest) < 2:
                    return "usage: /mem tag <id> <tags>"
                mem = tools.view({"memory_id": rest[0]})
                if "error" in mem:
                    return mem["error"]
                import os

                from fei_amd.memdir import utils as mu
                root = mu.get_memdir_base(tools.base)
                headers = dict(mem["headers"])
                headers["Tags"] = ",".join(rest[1:])
                path = os.path.join(mu.folder_path(mem["folder"], tools.base),
                                    mem["status"], mem["filename"])
                with open(path, "w", encoding="utf-8") as f:
                    f.write(mu.format_memory_content(headers, mem["content"]))
                return f"tagged {rest[0]}"
            if cmd == "server":
                from fei_amd.tools.memdir_connector import MemdirConnector
                conn = MemdirConnector(base=tools.base)
                sub = rest[0] if rest else "status"
                if sub == "start":
                    return str(conn.start_server_command())
                if sub == "stop":
                    return str(conn.stop_server_command())
                return str(conn.get_server_status())
            return f"unknown /mem command: {cmd} (try /mem help)"
        except Exception as e:  # noqa: BLE001 — UI surface, report not crash
            return f"error: {e}"


def main(provider: Optional[str] = None, model: Optional[str] = None,
         assistant=None) -> FeiChatApp:
    """Build the app; the caller runs it (reference: textual_chat.py:1044)."""
    if assistant is None:
        from fei_amd.core.assistant import Assistant
        from fei_amd.tools.code import create_code_tools
        from fei_amd.tools.memory_tools import create_memory_tools
        from fei_amd.tools.registry import ToolRegistry

        registry = ToolRegistry()
        create_code_tools(registry)
        create_memory_tools(registry)
        assistant = Assistant(provider=provider, model=model,
                              tool_registry=registry)
    return FeiChatApp(assistant=assistant)
