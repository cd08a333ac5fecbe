"""Cross-layer integration scenarios (scripted model, real tools, real
memdir, real MCP subprocess)."""

import sys
import textwrap

from fei_amd.core.assistant import Assistant
from fei_amd.core.backends import ScriptedBackend
from fei_amd.core.mcp import MCPClient, MCPManager, ProcessManager
from fei_amd.core.task_executor import TaskExecutor
from fei_amd.tools.code import create_code_tools
from fei_amd.tools.memory_tools import create_memory_tools
from fei_amd.tools.registry import ToolRegistry
from fei_amd.utils.config import Config


def _cfg(tmp_path):
    return Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)


def test_agent_stores_then_recalls_memory(tmp_path, memdir_base):
    """Turn 1: the agent persists a fact via memory_create. Turn 2 (fresh
    conversation): it recalls the fact via memory_search."""
    reg = ToolRegistry()
    create_memory_tools(reg, base=memdir_base)

    script = [
        {"tool_calls": [{"name": "memory_create",
                         "input": {"subject": "deploy key location",
                                   "body": "the key lives in vault/x9",
                                   "tags": "ops"}}]},
        {"content": "Stored it."},
        {"tool_calls": [{"name": "memory_search",
                         "input": {"query": "deploy key",
                                   "with_content": True}}]},
        {"content": "Found: the key lives in vault/x9 [TASK_COMPLETE]"},
    ]
    a = Assistant(config=_cfg(tmp_path), backend=ScriptedBackend(script),
                  tool_registry=reg)
    out1 = a.ask("remember where the deploy key is: vault/x9")
    assert "Stored" in out1
    a.reset()
    out2 = a.ask("where is the deploy key?")
    assert "vault/x9" in out2
    # the tool really hit the disk
    from fei_amd.memdir import utils as mu
    mems = mu.list_memories("", "new", include_content=True, base=memdir_base)
    assert mems and "vault/x9" in mems[0]["content"]


def test_agent_edits_code_and_validates(tmp_path):
    """Task flow: grep -> edit -> re-view, with the registry's validation
    rejecting a malformed call along the way."""
    target = tmp_path / "app.py"
    target.write_text("def add(a, b):\n    return a - b\n")
    reg = ToolRegistry()
    create_code_tools(reg)

    # TaskExecutor drives chat(): ONE tool round per iteration, and tool
    # calls in a continuation are ignored (reference parity) — so each
    # iteration is {tool round, then a text continuation}.
    script = [
        {"tool_calls": [{"name": "GrepTool",
                         "input": {"pattern": "return a", "path": str(tmp_path)}}]},
        {"content": "found the bug"},
        {"tool_calls": [{"name": "Edit",
                         "input": {"file_path": str(target),
                                   "old_string": "return a - b",
                                   "new_string": "return a + b"}}]},
        {"content": "edited"},
        {"tool_calls": [{"name": "Edit", "input": {"file_path": str(target)}}]},
        {"content": "fixed the bug [TASK_COMPLETE]"},
    ]
    a = Assistant(config=_cfg(tmp_path), backend=ScriptedBackend(script),
                  tool_registry=reg)
    result = TaskExecutor(a).execute_task("fix add()", max_iterations=8)
    assert result["complete"]
    assert target.read_text() == "def add(a, b):\n    return a + b\n"
    # the malformed third call produced an error tool_result, not a crash
    flat = str(a.conversation.messages)
    assert "missing required argument" in flat


def test_agent_with_mcp_tool_call(tmp_path):
    """The scripted model calls an mcp_<service>_<method> tool backed by a
    live stdio JSON-RPC subprocess."""
    server = tmp_path / "srv.py"
    server.write_text(textwrap.dedent("""
        import json, sys
        for line in sys.stdin:
            req = json.loads(line)
            resp = {"jsonrpc": "2.0", "id": req["id"],
                    "result": {"nodes": ["alpha", "beta"]}}
            sys.stdout.write(json.dumps(resp) + "\\n")
            sys.stdout.flush()
    """))
    client = MCPClient(process_manager=ProcessManager())
    client.add_server("kg", command=[sys.executable, str(server)])
    mgr = MCPManager(client=client)
    reg = ToolRegistry()
    mgr.attach_registry(reg)

    script = [
        {"tool_calls": [{"name": "mcp_kg_search_nodes",
                         "input": {"query": "alpha"}}]},
        {"content": "The graph has alpha and beta."},
    ]
    a = Assistant(config=_cfg(tmp_path), backend=ScriptedBackend(script),
                  tool_registry=reg, mcp_manager=mgr)
    out = a.ask("what nodes exist?")
    assert "alpha" in out
    flat = str(a.conversation.messages)
    assert "beta" in flat
    mgr.shutdown()
