"""Tool registry: name -> handler map with JSON-schema argument validation.

Parity target: the reference ToolRegistry (fei/tools/registry.py:92-153
validation, :250-297 dispatch, :299-338 async trampoline, :340-467 MCP
special-casing). Re-designed: validation is a small self-contained JSON-schema
subset checker (type/required/enum/properties), dispatch is synchronous with
an asyncio trampoline for coroutine handlers, and MCP tools are handled by a
pluggable prefix hook rather than hard-coded names.
"""

from __future__ import annotations

import asyncio
import inspect
import threading
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from fei_amd.utils.logging import get_logger

logger = get_logger("tools.registry")


class ToolValidationError(ValueError):
    """Raised when tool arguments do not match the declared schema."""


def _check_type(value: Any, expected: str) -> bool:
    if expected == "string":
        return isinstance(value, str)
    if expected == "integer":
        return isinstance(value, int) and not isinstance(value, bool)
    if expected == "number":
        return isinstance(value, (int, float)) and not isinstance(value, bool)
    if expected == "boolean":
        return isinstance(value, bool)
    if expected == "array":
        return isinstance(value, list)
    if expected == "object":
        return isinstance(value, dict)
    if expected == "null":
        return value is None
    return True


def validate_schema(args: Dict[str, Any], schema: Dict[str, Any], path: str = "") -> List[str]:
    """Validate ``args`` against a JSON-schema subset. Returns error strings."""
    errors: List[str] = []
    if not isinstance(schema, dict):
        return errors
    required = schema.get("required", [])
    props = schema.get("properties", {})
    for name in required:
        if name not in args:
            errors.append(f"missing required argument {path}{name!r}")
    for name, value in args.items():
        prop = props.get(name)
        if prop is None:
            continue
        expected = prop.get("type")
        if isinstance(expected, list):
            if not any(_check_type(value, t) for t in expected):
                errors.append(f"argument {path}{name!r} has wrong type (expected one of {expected})")
        elif isinstance(expected, str) and not _check_type(value, expected):
            errors.append(f"argument {path}{name!r} has wrong type (expected {expected})")
        if "enum" in prop and value not in prop["enum"]:
            errors.append(f"argument {path}{name!r} not in enum {prop['enum']}")
        if expected == "object" and isinstance(value, dict):
            errors.extend(validate_schema(value, prop, path=f"{path}{name}."))
        if expected == "array" and isinstance(value, list) and "items" in prop:
            item_schema = prop["items"]
            item_type = item_schema.get("type")
            for i, item in enumerate(value):
                if isinstance(item_type, str) and not _check_type(item, item_type):
                    errors.append(f"argument {path}{name}[{i}] has wrong type (expected {item_type})")
    return errors


@dataclass
class Tool:
    name: str
    description: str
    input_schema: Dict[str, Any]
    handler: Callable[..., Any]
    tags: List[str] = field(default_factory=list)

    def validate_arguments(self, args: Dict[str, Any]) -> List[str]:
        return validate_schema(args, self.input_schema)

    def to_schema(self) -> Dict[str, Any]:
        """Anthropic-style tool spec dict."""
        return {
            "name": self.name,
            "description": self.description,
            "input_schema": self.input_schema,
        }


class ToolRegistry:
    """Thread-safe name -> Tool map with validated dispatch."""

    def __init__(self):
        self._tools: Dict[str, Tool] = {}
        self._lock = threading.RLock()
        # Hooks by name-prefix: e.g. "mcp_" -> callable(name, args) -> result.
        self._prefix_hooks: Dict[str, Callable[[str, Dict[str, Any]], Any]] = {}

    # -- registration --------------------------------------------------------

    def register_tool(
        self,
        name: str,
        description: str,
        input_schema: Dict[str, Any],
        handler: Callable[..., Any],
        tags: Optional[List[str]] = None,
    ) -> Tool:
        tool = Tool(name, description, input_schema, handler, tags or [])
        with self._lock:
            if name in self._tools:
                logger.warning("re-registering tool %s", name)
            self._tools[name] = tool
        return tool

    def register(self, tool: Tool) -> Tool:
        with self._lock:
            self._tools[tool.name] = tool
        return tool

    def register_prefix_hook(self, prefix: str, hook: Callable[[str, Dict[str, Any]], Any]) -> None:
        """Dispatch any un-registered tool whose name starts with ``prefix``
        to ``hook`` (used by the MCP layer: mcp_<service>_<method> names)."""
        with self._lock:
            self._prefix_hooks[prefix] = hook

    def unregister(self, name: str) -> bool:
        with self._lock:
            return self._tools.pop(name, None) is not None

    # -- queries -------------------------------------------------------------

    def get_tool(self, name: str) -> Optional[Tool]:
        with self._lock:
            return self._tools.get(name)

    def list_tools(self) -> List[str]:
        with self._lock:
            return sorted(self._tools)

    def get_schemas(self) -> List[Dict[str, Any]]:
        with self._lock:
            return [t.to_schema() for t in self._tools.values()]

    # -- dispatch ------------------------------------------------------------

    def execute_tool(self, name: str, args: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
        """Validate and run a tool. Always returns a dict; on failure the dict
        has an ``error`` key (reference behavior: registry.py:250-297)."""
        args = args or {}
        with self._lock:
            tool = self._tools.get(name)
            hooks = dict(self._prefix_hooks)
        if tool is None:
            for prefix, hook in hooks.items():
                if name.startswith(prefix):
                    try:
                        return self._as_result(hook(name, args))
                    except Exception as e:  # noqa: BLE001 - tool errors go to the LLM
                        logger.exception("prefix hook for %s failed", name)
                        return {"error": f"{type(e).__name__}: {e}"}
            return {"error": f"unknown tool: {name}"}

        errors = tool.validate_arguments(args)
        if errors:
            return {"error": "invalid arguments: " + "; ".join(errors)}

        try:
            result = tool.handler(args)
            if inspect.iscoroutine(result):
                result = self._run_coroutine(result)
            return self._as_result(result)
        except Exception as e:  # noqa: BLE001 - tool errors are reported, not raised
            logger.exception("tool %s failed", name)
            return {"error": f"{type(e).__name__}: {e}"}

    @staticmethod
    def _as_result(result: Any) -> Dict[str, Any]:
        if isinstance(result, dict):
            return result
        return {"result": result}

    @staticmethod
    def _run_coroutine(coro) -> Any:
        """Run a coroutine handler to completion even when called from inside
        a running event loop (reference trampoline: registry.py:299-338)."""
        try:
            asyncio.get_running_loop()
        except RuntimeError:
            return asyncio.run(coro)
        # Called from within a loop: run in a fresh loop on a worker thread.
        out: Dict[str, Any] = {}

        def _runner():
            try:
                out["result"] = asyncio.run(coro)
            except Exception as e:  # noqa: BLE001
                out["error"] = e

        t = threading.Thread(target=_runner, daemon=True)
        t.start()
        t.join()
        if "error" in out:
            raise out["error"]
        return out.get("result")
