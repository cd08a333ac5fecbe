"""ToolRegistry validation + dispatch tests."""

import asyncio

from fei_amd.tools.registry import ToolRegistry, validate_schema


SCHEMA = {
    "type": "object",
    "properties": {
        "name": {"type": "string"},
        "count": {"type": "integer"},
        "mode": {"type": "string", "enum": ["a", "b"]},
        "items": {"type": "array", "items": {"type": "string"}},
    },
    "required": ["name"],
}


def test_validate_ok():
    assert validate_schema({"name": "x", "count": 2, "mode": "a"}, SCHEMA) == []


def test_validate_missing_required():
    errs = validate_schema({}, SCHEMA)
    assert any("name" in e for e in errs)


def test_validate_wrong_type_and_enum():
    errs = validate_schema({"name": 5, "mode": "z", "items": ["a", 1]}, SCHEMA)
    assert len(errs) == 3


def test_registry_dispatch():
    reg = ToolRegistry()
    reg.register_tool("Echo", "echo", SCHEMA, lambda args: {"echo": args["name"]})
    assert reg.execute_tool("Echo", {"name": "hi"}) == {"echo": "hi"}
    out = reg.execute_tool("Echo", {})
    assert "error" in out
    out = reg.execute_tool("Nope", {"name": "hi"})
    assert "unknown tool" in out["error"]


def test_registry_handler_exception_reported():
    reg = ToolRegistry()

    def boom(args):
        raise RuntimeError("bad")

    reg.register_tool("Boom", "boom", {"type": "object", "properties": {}}, boom)
    out = reg.execute_tool("Boom", {})
    assert "RuntimeError" in out["error"]


def test_registry_async_handler():
    reg = ToolRegistry()

    async def ahandler(args):
        await asyncio.sleep(0)
        return {"ok": True}

    reg.register_tool("Async", "async", {"type": "object", "properties": {}}, ahandler)
    assert reg.execute_tool("Async", {}) == {"ok": True}


def test_registry_prefix_hook():
    reg = ToolRegistry()
    reg.register_prefix_hook("mcp_", lambda name, args: {"routed": name})
    assert reg.execute_tool("mcp_memory_read_graph", {}) == {"routed": "mcp_memory_read_graph"}


def test_code_tools_registered():
    from fei_amd.tools.code import create_code_tools
    reg = ToolRegistry()
    create_code_tools(reg)
    assert len(reg.list_tools()) == 14
    for name in ["GlobTool", "GrepTool", "View", "Edit", "Replace", "LS",
                 "RegexEdit", "BatchGlob", "FindInFiles", "SmartSearch",
                 "RepoMap", "RepoSummary", "RepoDependencies", "Shell"]:
        assert name in reg.list_tools()
