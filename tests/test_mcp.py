"""MCP client tests with a real local stdio JSON-RPC echo server
(the reference mocked Popen; a live subprocess covers more —
fei/tests/test_mcp.py:42-96 is the parity target)."""

import sys
import textwrap

import pytest

from fei_amd.core.mcp import MCPClient, MCPError, MCPManager, ProcessManager
from fei_amd.tools.registry import ToolRegistry

ECHO_SERVER = textwrap.dedent("""
    import json, sys
    for line in sys.stdin:
        try:
            req = json.loads(line)
        except Exception:
            continue
        method = req.get("method")
        if method == "boom":
            resp = {"jsonrpc": "2.0", "id": req["id"],
                    "error": {"code": -1, "message": "kaboom"}}
        else:
            resp = {"jsonrpc": "2.0", "id": req["id"],
                    "result": {"method": method, "params": req.get("params")}}
        sys.stdout.write(json.dumps(resp) + "\\n")
        sys.stdout.flush()
""")


@pytest.fixture
def client(tmp_path):
    script = tmp_path / "echo_server.py"
    script.write_text(ECHO_SERVER)
    c = MCPClient(process_manager=ProcessManager())
    c.add_server("echo", command=[sys.executable, str(script)])
    yield c
    c.procs.stop_all()


def test_stdio_roundtrip(client):
    result = client.call_service("echo", "create_entities",
                                 {"entities": [{"name": "x"}]})
    assert result["method"] == "create_entities"
    assert result["params"]["entities"][0]["name"] == "x"


def test_stdio_error_raises(client):
    with pytest.raises(MCPError, match="kaboom"):
        client.call_service("echo", "boom", {})


def test_stdio_sequential_ids(client):
    for i in range(3):
        out = client.call_service("echo", f"m{i}", {})
        assert out["method"] == f"m{i}"


def test_unknown_server(client):
    with pytest.raises(MCPError, match="unknown"):
        client.call_service("nope", "m", {})


def test_process_stop_and_restart(client):
    client.call_service("echo", "warm", {})
    assert client.stop_server("echo") is True
    # restarts transparently on next call
    out = client.call_service("echo", "again", {})
    assert out["method"] == "again"


def test_env_server_config(monkeypatch, tmp_path):
    script = tmp_path / "s.py"
    script.write_text(ECHO_SERVER)
    monkeypatch.setenv("FEI_MCP_SERVER_MYTOOL",
                       f"stdio:{sys.executable} {script}")
    monkeypatch.setenv("FEI_MCP_SERVER_WEB", "http://localhost:9/rpc")
    monkeypatch.setenv("FEI_MCP_SERVER_BAD", "not-a-url")
    c = MCPClient(process_manager=ProcessManager())
    assert "mytool" in c.list_servers()
    assert "web" in c.list_servers()
    assert "bad" not in c.list_servers()
    c.procs.stop_all()


def test_manager_registry_hook(client):
    mgr = MCPManager(client=client)
    reg = ToolRegistry()
    mgr.attach_registry(reg)
    out = reg.execute_tool("mcp_echo_search_nodes", {"query": "q"})
    assert out["result"]["method"] == "search_nodes"
    out = reg.execute_tool("mcp_missing_method", {})
    assert "error" in out


def test_memory_service_facade(client):
    mgr = MCPManager(client=client)
    mgr.memory.server = "echo"
    out = mgr.memory.search_nodes("hello")
    assert out["params"]["query"] == "hello"


def test_stdio_server_restart_after_crash(tmp_path):
    """A stdio server that dies mid-conversation is restarted once and the
    call retried (stateless-server recovery)."""
    import sys
    import textwrap
    from fei_amd.core.mcp import MCPClient, ProcessManager
    flag = tmp_path / "started_once"
    server = tmp_path / "crashy.py"
    server.write_text(textwrap.dedent(f"""
        import json, os, sys
        flag = {str(flag)!r}
        if not os.path.exists(flag):
            open(flag, "w").close()
            sys.exit(1)            # crash on first launch
        for line in sys.stdin:
            req = json.loads(line)
            sys.stdout.write(json.dumps({{"jsonrpc": "2.0", "id": req["id"],
                                          "result": {{"ok": True}}}}) + "\\n")
            sys.stdout.flush()
    """))
    client = MCPClient(process_manager=ProcessManager())
    client.add_server("crashy", command=[sys.executable, str(server)])
    out = client.call_service("crashy", "ping", {})
    assert out == {"ok": True}
    client.stop_server("crashy")
