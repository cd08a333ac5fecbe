"""Textual TUI: chat app with the /mem command suite.

Parity: reference fei/ui/textual_chat.py (1,070 LoC): FeiChatApp with
message panels, a memory-command suggester, the /mem command set
(help/list/search/view/save/tag/server — :557-970) calling the memory
tool handlers directly, background assistant processing (:1002-1031), and
``main()`` returning the App for the caller to ``.run()`` (:1044-1062).
"""

from __future__ import annotations

from typing import Optional

from textual.app import App, ComposeResult
from textual.containers import VerticalScroll
from textual.suggester import Suggester
from textual.widgets import Footer, Header, Input, Static

MEM_COMMANDS = [
    "/mem help", "/mem list", "/mem search ", "/mem view ", "/mem save ",
    "/mem tag ", "/mem server start", "/mem server stop", "/mem server status",
    "/mem index", "/mem semantic ", "/mem keyword ",
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


class MemCommandSuggester(Suggester):
    """Autocomplete for /mem commands (reference: textual_chat.py:119-229)."""

    def __init__(self):
        super().__init__(use_cache=False, case_sensitive=False)

    async def get_suggestion(self, value: str) -> Optional[str]:
        if not value.startswith("/"):
            return None
        for cmd in MEM_COMMANDS:
            if cmd.startswith(value) and cmd != value:
                return cmd
        return None


class ChatMessage(Static):
    pass


class FeiChatApp(App):
    CSS = """
    #chat { height: 1fr; }
    ChatMessage { padding: 0 1; margin-bottom: 1; }
    .user { background: $boost; }
    .assistant { }
    .system { color: $text-muted; }
    """
    BINDINGS = [("ctrl+c", "quit", "Quit"), ("ctrl+l", "clear", "Clear")]

    def __init__(self, assistant=None, memory_tools=None, **kwargs):
        super().__init__(**kwargs)
        self.assistant = assistant
        self.memory_tools = memory_tools

    # -- layout --------------------------------------------------------------

    def compose(self) -> ComposeResult:
        yield Header(show_clock=True)
        yield VerticalScroll(id="chat")
        yield Input(placeholder="Message (or /mem ...)",
                    suggester=MemCommandSuggester(), id="input")
        yield Footer()

    def add_message(self, text: str, kind: str = "assistant") -> None:
        chat = self.query_one("#chat", VerticalScroll)
        prefix = {"user": "you> ", "assistant": "fei> ", "system": ""}[kind]
        msg = ChatMessage(prefix + text)
        msg.add_class(kind)
        chat.mount(msg)
        chat.scroll_end(animate=False)

    def action_clear(self) -> None:
        if self.assistant is not None:
            self.assistant.reset()
        self.query_one("#chat", VerticalScroll).remove_children()

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
        self.run_worker(self.process_with_assistant(text), exclusive=True)

    async def process_with_assistant(self, text: str) -> None:
        """Background turn (reference: textual_chat.py:1002-1031)."""
        if self.assistant is None:
            self.add_message("(no assistant configured)", "system")
            return
        import asyncio
        loop = asyncio.get_running_loop()
        response = await loop.run_in_executor(None, self.assistant.ask, text)
        self.add_message(response or "(no response)")

    # -- /mem commands (reference: textual_chat.py:557-970) -------------------

    def handle_memory_command(self, text: str) -> str:
        tools = self.memory_tools
        if tools is None:
            from fei_amd.tools.memory_tools import MemoryTools
            tools = self.memory_tools = MemoryTools()
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
                from fei_amd.memdir import utils as mu
                import os
                root = mu.get_memdir_base(tools.base)
                headers = dict(mem["headers"])
                headers["Tags"] = ",".join(rest[1:])
                path = os.path.join(root, mem["folder"], mem["status"],
                                    mem["filename"]) if mem["folder"] else \
                    os.path.join(root, mem["status"], mem["filename"])
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
