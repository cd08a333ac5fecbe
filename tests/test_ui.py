"""UI tests: CLI chat loop / single message with the scripted backend,
history persistence, and TUI construction without running the app
(reference parity: test_textual.py:20-31)."""

import io
import json
import os

import pytest


@pytest.fixture
def home(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))
    # ChatHistory resolves ~ at import time; patch the path constant
    import fei_amd.ui.cli as cli_mod
    monkeypatch.setattr(cli_mod, "HISTORY_PATH",
                        str(tmp_path / ".fei" / "history.json"))
    return tmp_path


def test_single_message_scripted(home):
    from fei_amd.ui.cli import CLI
    cli = CLI(script=[{"content": "forty-two"}], with_memory=False)
    out = cli.single_message("meaning of life?")
    assert out == "forty-two"
    # history persisted
    data = json.loads((home / ".fei" / "history.json").read_text())
    assert data[0]["prompt"] == "meaning of life?"


def test_task_mode(home):
    from fei_amd.ui.cli import CLI
    cli = CLI(script=[{"content": "step"}, {"content": "done [TASK_COMPLETE]"}],
              with_memory=False)
    out = cli.single_message("do it", task=True)
    assert out == "done"


def test_chat_loop_commands(home):
    from fei_amd.ui.cli import CLI
    cli = CLI(script=[{"content": "hi there"}], with_memory=False)
    stdin = io.StringIO("hello\nhistory\nexit\n")
    stdout = io.StringIO()
    rc = cli.chat_loop(stdin=stdin, stdout=stdout)
    assert rc == 0
    text = stdout.getvalue()
    assert "fei> hi there" in text
    assert "> hello" in text          # history echo


def test_chat_loop_clear(home):
    from fei_amd.ui.cli import CLI
    cli = CLI(script=[{"content": "x"}], with_memory=False)
    stdin = io.StringIO("one\nclear\nquit\n")
    stdout = io.StringIO()
    cli.chat_loop(stdin=stdin, stdout=stdout)
    assert cli.assistant.conversation.messages == []


def test_history_cap(home):
    from fei_amd.ui.cli import ChatHistory
    h = ChatHistory(str(home / ".fei" / "history.json"))
    for i in range(120):
        h.add(f"p{i}", f"r{i}")
    assert len(h.entries) == 100
    assert h.entries[0]["prompt"] == "p20"


def test_cli_main_message(home, capsys):
    from fei_amd.ui.cli import main
    rc = main(["--provider", "stub", "--no-memory", "-m", "echo this back"])
    assert rc == 0
    assert "echo this back" in capsys.readouterr().out


def test_cli_search_subcommand(home, tmp_path, monkeypatch, capsys):
    monkeypatch.setenv("MEMDIR_BASE", str(tmp_path / "Memdir"))
    from fei_amd.memdir import utils as mu
    mu.create_memory("", {"Subject": "findme note"}, "", status="cur")
    from fei_amd.ui.cli import main
    rc = main(["search", "findme"])
    assert rc == 0
    assert "findme note" in capsys.readouterr().out


def test_tui_construction(home):
    """Construct the app + exercise /mem handlers without running it."""
    from fei_amd.tools.memory_tools import MemoryTools
    from fei_amd.ui.tui import FeiChatApp

    tools = MemoryTools(base=str(home / "Memdir"))
    tools.create({"subject": "tui memory", "tags": "t"})
    app = FeiChatApp(assistant=None, memory_tools=tools)
    out = app.handle_memory_command("/mem list")
    assert "tui memory" in out
    out = app.handle_memory_command("/mem search tui")
    assert "tui memory" in out
    out = app.handle_memory_command("/mem help")
    assert "/mem save" in out
    out = app.handle_memory_command("/mem bogus")
    assert "unknown" in out


def test_tui_suggester():
    import asyncio
    from fei_amd.ui.tui import MemCommandSuggester
    s = MemCommandSuggester()
    assert asyncio.run(s.get_suggestion("/mem se")) == "/mem search "
    assert asyncio.run(s.get_suggestion("hello")) is None


def test_doctor(home, capsys):
    from fei_amd.ui.cli import main
    rc = main(["doctor"])
    out = capsys.readouterr().out
    assert rc == 0
    assert "fei_amd" in out and "torch" in out and "kernel lib" in out


def test_cli_local_engine_end_to_end(home, capsys):
    """Full local path: CLI -> Assistant -> LocalBackend -> LocalEngine
    (tiny model on CPU; output is random-weights text but the whole agent
    stack runs for real)."""
    from fei_amd.ui.cli import main
    rc = main(["--provider", "local", "--model", "llama3-tiny",
               "--no-memory", "-m", "hello"])
    assert rc == 0
    assert capsys.readouterr().out.strip()


def test_history_flags(home, capsys):
    from fei_amd.ui.cli import ChatHistory, main
    h = ChatHistory()
    h.add("first prompt", "first reply")
    h.add("second prompt", "second reply")
    assert main(["history", "--limit", "1"]) == 0
    out = capsys.readouterr().out
    assert "second prompt" in out and "first prompt" not in out
    assert main(["history", "--load", "0"]) == 0
    assert "first reply" in capsys.readouterr().out
    assert main(["history", "--clear"]) == 0
    capsys.readouterr()
    main(["history"])
    assert "prompt" not in capsys.readouterr().out
    assert main(["history", "--load", "5"]) == 1


def test_cli_task_mode_local_engine(home, capsys):
    """--task through the REAL local engine (tiny): random weights emit no
    valid tool calls, so the loop must hit the iteration cap and exit
    cleanly rather than crash on garbage output."""
    from fei_amd.ui.cli import main
    rc = main(["--provider", "local", "--model", "llama3-tiny",
               "--no-memory", "--task", "do something",
               "--max-iterations", "2"])
    assert rc == 0
    assert capsys.readouterr().out is not None


def test_tui_mem_keyword(home):
    from fei_amd.ui.tui import FeiChatApp
    from fei_amd.tools.memory_tools import MemoryTools
    tools = MemoryTools(base=str(home / "Memdir"))
    tools.create({"subject": "keyword target", "body": "findable giraffe"})
    tools.index_build({})
    app = FeiChatApp(assistant=None, memory_tools=tools)
    out = app.handle_memory_command("/mem keyword giraffe")
    assert "keyword target" in out


# -- TUI depth (reference textual_chat.py:48-229, 557-970) -------------------

def test_tui_pilot_user_message_and_mem(home):
    """Run the app headless with Textual's Pilot: submitting input mounts a
    UserMessage panel; a /mem command mounts its SystemMessage reply."""
    import asyncio

    from fei_amd.tools.memory_tools import MemoryTools
    from fei_amd.ui.tui import (FeiChatApp, SystemMessage, UserMessage)

    tools = MemoryTools(base=str(home / "Memdir"))
    tools.create({"subject": "pilot memory", "tags": "t"})

    async def run():
        app = FeiChatApp(assistant=None, memory_tools=tools)
        async with app.run_test() as pilot:
            inp = app.query_one("#input")
            inp.value = "/mem list"
            await pilot.press("enter")
            await pilot.pause()
            users = app.query(UserMessage)
            systems = app.query(SystemMessage)
            assert len(users) == 1
            assert any("pilot memory" in str(m.content) for m in systems)

    asyncio.run(run())


def test_tui_pilot_slash_commands(home):
    import asyncio

    from fei_amd.ui.tui import FeiChatApp, StatusPanel, SystemMessage

    async def run():
        app = FeiChatApp(assistant=None)
        async with app.run_test() as pilot:
            inp = app.query_one("#input")
            inp.value = "/help"
            await pilot.press("enter")
            await pilot.pause()
            assert any("/mem save" in str(m.content)
                       for m in app.query(SystemMessage))
            # F2 toggles the status sidebar
            panel = app.query_one("#status", StatusPanel)
            assert not panel.has_class("visible")
            await pilot.press("f2")
            assert panel.has_class("visible")
            # /clear empties the chat
            inp.value = "/clear"
            await pilot.press("enter")
            await pilot.pause()
            texts = [str(m.content) for m in app.query(SystemMessage)]
            assert "(cleared)" in texts

    asyncio.run(run())


def test_tui_pilot_assistant_turn_with_tool_panels(home):
    """A full background assistant turn through the stub backend renders
    the assistant panel (and tool panels when the turn ran tools)."""
    import asyncio

    from fei_amd.core.assistant import Assistant
    from fei_amd.tools.code import create_code_tools
    from fei_amd.tools.registry import ToolRegistry
    from fei_amd.ui.tui import AssistantMessage, FeiChatApp

    registry = ToolRegistry()
    create_code_tools(registry)
    assistant = Assistant(provider="stub", tool_registry=registry)

    async def run():
        app = FeiChatApp(assistant=assistant)
        async with app.run_test() as pilot:
            inp = app.query_one("#input")
            inp.value = "hello there"
            await pilot.press("enter")
            for _ in range(50):
                await pilot.pause(0.1)
                if app.query(AssistantMessage):
                    break
            msgs = app.query(AssistantMessage)
            assert len(msgs) == 1
            assert "hello there" in str(msgs.first().content)

    asyncio.run(run())


def test_tui_suggester_completes_slash_and_ids(home):
    import asyncio

    from fei_amd.tools.memory_tools import MemoryTools
    from fei_amd.ui.tui import CommandSuggester, FeiChatApp

    tools = MemoryTools(base=str(home / "Memdir"))
    out = tools.create({"subject": "sugg", "tags": ""})
    mid = out["memory_id"]
    app = FeiChatApp(assistant=None, memory_tools=tools)
    s = CommandSuggester(app)
    assert asyncio.run(s.get_suggestion("/he")) == "/help"
    assert asyncio.run(s.get_suggestion("/mem se")) == "/mem search "
    got = asyncio.run(s.get_suggestion("/mem view " + mid[:3]))
    assert got == "/mem view " + mid


def test_tui_history_and_stats_commands(home):
    from fei_amd.core.assistant import Assistant
    from fei_amd.ui.tui import FeiChatApp

    assistant = Assistant(provider="stub")
    assistant.chat("first message")
    app = FeiChatApp(assistant=assistant)
    out = app.handle_slash_command("/history")
    assert "first message" in out
    out = app.handle_slash_command("/stats")
    assert "turn 0" in out
