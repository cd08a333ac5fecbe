"""MCP integration (reference: examples/mcp_brave_search.py, no network
needed): a stdio JSON-RPC server becomes mcp_<name>_<method> tools."""
import json, os, sys, textwrap, tempfile
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.core.mcp import MCPClient, MCPManager, ProcessManager
from fei_amd.tools.registry import ToolRegistry

with tempfile.TemporaryDirectory() as d:
    server = os.path.join(d, "kv.py")
    with open(server, "w") as f:
        f.write(textwrap.dedent("""
            import json, sys
            store = {}
            for line in sys.stdin:
                req = json.loads(line)
                p = req.get("params", {})
                if req["method"] == "set":
                    store[p["key"]] = p["value"]; result = {"ok": True}
                else:
                    result = {"value": store.get(p.get("key"))}
                sys.stdout.write(json.dumps({"jsonrpc": "2.0",
                                             "id": req["id"],
                                             "result": result}) + "\\n")
                sys.stdout.flush()
        """))
    client = MCPClient(process_manager=ProcessManager())
    client.add_server("kv", command=[sys.executable, server])
    mgr = MCPManager(client=client)
    reg = ToolRegistry()
    mgr.attach_registry(reg)
    print(reg.execute_tool("mcp_kv_set", {"key": "gpu", "value": "MI355X"}))
    print(reg.execute_tool("mcp_kv_get", {"key": "gpu"}))
    mgr.shutdown()
