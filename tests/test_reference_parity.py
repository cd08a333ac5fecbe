"""Mechanical parity checks against the reference tree (skipped when
/root/reference is absent). These encode the 'a reference user finds
everything they need' contract: tool names, argument names, CLI
subcommands and HTTP routes must be accepted verbatim."""

import ast
import os
import re

import pytest

REF = "/root/reference"

pytestmark = pytest.mark.skipif(not os.path.isdir(REF),
                                reason="reference tree not mounted")


def _ref_tool_schemas():
    src = open(os.path.join(REF, "fei/tools/definitions.py")).read()
    out = {}
    for m in re.finditer(r'^[A-Z_]+\s*=\s*{', src, re.M):
        i = src.index('{', m.start())
        depth = 0
        for j in range(i, len(src)):
            if src[j] == '{':
                depth += 1
            elif src[j] == '}':
                depth -= 1
                if depth == 0:
                    break
        try:
            d = ast.literal_eval(src[i:j + 1])
        except (ValueError, SyntaxError):
            continue
        if isinstance(d, dict) and "name" in d and "input_schema" in d:
            out[d["name"]] = set(d["input_schema"].get("properties", {}))
    return out


def _our_tool_schemas():
    import fei_amd.tools.definitions as d
    out = {}
    for name in dir(d):
        v = getattr(d, name)
        if isinstance(v, dict) and "name" in v and "input_schema" in v:
            out[v["name"]] = set(v["input_schema"].get("properties", {}))
    return out


def test_tool_schemas_are_superset():
    ref, ours = _ref_tool_schemas(), _our_tool_schemas()
    missing_tools = [n for n in ref if n not in ours]
    assert not missing_tools, f"tools missing vs reference: {missing_tools}"
    gaps = {n: sorted(ref[n] - ours[n]) for n in ref if ref[n] - ours[n]}
    assert not gaps, f"argument names missing vs reference: {gaps}"


def _subcommands(path):
    src = open(path).read()
    return set(re.findall(r'add_parser\(["\']([a-z-]+)["\']', src))


def test_memdir_cli_subcommands():
    ref = _subcommands(os.path.join(REF, "memdir_tools/cli.py"))
    ours = _subcommands("fei_amd/memdir/cli.py")
    assert ref <= ours, f"memdir CLI missing: {sorted(ref - ours)}"


def test_memorychain_cli_subcommands():
    ref = _subcommands(os.path.join(REF, "memdir_tools/memorychain_cli.py"))
    ours = _subcommands("fei_amd/memorychain/cli.py")
    assert ref <= ours, f"memorychain CLI missing: {sorted(ref - ours)}"


def _routes(path, pattern):
    src = open(path).read()
    return set(re.findall(pattern, src))


def test_memorychain_node_routes():
    ref = _routes(os.path.join(REF, "memdir_tools/memorychain.py"),
                  r"route\(['\"](/memorychain/[^'\"<]*)")
    ours = _routes("fei_amd/memorychain/node.py",
                   r'@app\.(?:get|post|route)\("(/memorychain/[^"<]*)')
    # normalize: drop trailing path params
    norm = lambda s: {r.rstrip("/") for r in s}
    missing = norm(ref) - norm(ours)
    assert not missing, f"node routes missing: {sorted(missing)}"


def test_memdir_server_routes():
    ref = _routes(os.path.join(REF, "memdir_tools/server.py"),
                  r"route\(['\"]([^'\"<]*)")
    ours = _routes("fei_amd/memdir/server.py",
                   r'@app\.(?:get|post|put|delete|route)\("([^"<]*)')
    norm = lambda s: {r.rstrip("/") for r in s if r.startswith("/")}
    missing = norm(ref) - norm(ours)
    assert not missing, f"server routes missing: {sorted(missing)}"


def test_memory_tool_schemas_are_superset():
    src = open(os.path.join(REF, "fei/tools/memory_tools.py")).read()
    m = re.search(r'MEMORY_TOOL_SCHEMAS\s*=\s*{', src)
    i = src.index('{', m.start())
    depth = 0
    for j in range(i, len(src)):
        if src[j] == '{':
            depth += 1
        elif src[j] == '}':
            depth -= 1
            if depth == 0:
                break
    ref = {k: set(v.get("properties", {}))
           for k, v in ast.literal_eval(src[i:j + 1]).items()}
    import fei_amd.tools.memory_tools as mt
    ours = {}
    for name in dir(mt):
        v = getattr(mt, name)
        if isinstance(v, dict) and "name" in v and "input_schema" in v:
            ours[v["name"]] = set(v["input_schema"].get("properties", {}))
        elif isinstance(v, list):                      # SERVER_TOOLS
            for t in v:
                if isinstance(t, dict) and "name" in t:
                    ours[t["name"]] = set(
                        t.get("input_schema", {}).get("properties", {}))
    gaps = {}
    for n, fields in ref.items():
        if n not in ours:
            gaps[n] = "tool missing"
        elif fields - ours[n]:
            gaps[n] = sorted(fields - ours[n])
    assert not gaps, f"memory tool gaps: {gaps}"
