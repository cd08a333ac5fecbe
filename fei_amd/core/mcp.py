"""MCP (Model Context Protocol) client layer.

Parity: reference fei/core/mcp.py (1,184 LoC):
  - ProcessManager: start stdio servers in their own process group,
    SIGTERM -> SIGKILL stop, atexit cleanup (mcp.py:40-191)
  - MCPClient: server config from config + ``FEI_MCP_SERVER_<NAME>`` env
    vars (mcp.py:242-298), URL validation (:300-323), stdio JSON-RPC 2.0
    with a poll loop (:553-628), HTTP JSON-RPC (:658-716)
  - service facades: memory (entity/relation/observation graph ops,
    :753-864), fetch (:867), brave_search with a direct-API fallback
    (:911-1010), github (:1045)
  - MCPManager exposing ``.memory/.fetch/.brave_search/.github``
    (:1097-1185) and wiring ``mcp_<service>_<method>`` tool names into the
    registry (registry.py:340-467 — here via the registry prefix hook).

Note: this image has no network egress; HTTP servers and npx-spawned
servers only work where the environment provides them. The stdio protocol
path is fully functional and covered by tests with a local echo server.
"""

from __future__ import annotations

import atexit
import json
import os
import shlex
import signal
import subprocess
import threading
import time
from typing import Any, Dict, List, Optional

import requests

from fei_amd.utils.config import get_config
from fei_amd.utils.logging import get_logger

logger = get_logger("core.mcp")

STDIO_TIMEOUT_S = 30.0


class ProcessManager:
    """Own stdio server subprocesses; group-kill on stop/exit."""

    def __init__(self):
        self._procs: Dict[str, subprocess.Popen] = {}
        self._lock = threading.Lock()
        atexit.register(self.stop_all)

    def start(self, name: str, command: List[str],
              env: Optional[Dict[str, str]] = None) -> subprocess.Popen:
        with self._lock:
            proc = self._procs.get(name)
            if proc and proc.poll() is None:
                return proc
            proc = subprocess.Popen(
                command,
                stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                stderr=subprocess.DEVNULL,
                env={**os.environ, **(env or {})},
                start_new_session=True,        # own process group
                text=True, bufsize=1,
            )
            self._procs[name] = proc
            return proc

    def stop(self, name: str, grace_s: float = 3.0) -> bool:
        with self._lock:
            proc = self._procs.pop(name, None)
        if proc is None or proc.poll() is not None:
            return False
        try:
            os.killpg(os.getpgid(proc.pid), signal.SIGTERM)
            try:
                proc.wait(timeout=grace_s)
            except subprocess.TimeoutExpired:
                os.killpg(os.getpgid(proc.pid), signal.SIGKILL)
        except (ProcessLookupError, PermissionError):
            pass
        return True

    def stop_all(self) -> None:
        for name in list(self._procs):
            self.stop(name)

    def get(self, name: str) -> Optional[subprocess.Popen]:
        with self._lock:
            return self._procs.get(name)


class MCPError(RuntimeError):
    pass


class MCPClient:
    """JSON-RPC 2.0 over stdio subprocesses or HTTP endpoints."""

    def __init__(self, config=None, process_manager: Optional[ProcessManager] = None):
        self.config = config or get_config()
        self.procs = process_manager or ProcessManager()
        self.servers: Dict[str, Dict[str, Any]] = {}
        self._id = 0
        self._lock = threading.Lock()
        self._load_server_configs()

    # -- config --------------------------------------------------------------

    def _load_server_configs(self) -> None:
        """Env convention (reference: mcp.py:242-298):
        ``FEI_MCP_SERVER_<NAME>=stdio:<command>`` or ``...=<http url>``."""
        for key, value in os.environ.items():
            if not key.startswith("FEI_MCP_SERVER_"):
                continue
            name = key[len("FEI_MCP_SERVER_"):].lower()
            if value.startswith("stdio:"):
                self.servers[name] = {"type": "stdio",
                                      "command": shlex.split(value[6:])}
            elif self._valid_url(value):
                self.servers[name] = {"type": "http", "url": value}
            else:
                logger.warning("ignoring invalid MCP server config %s", key)

    @staticmethod
    def _valid_url(url: str) -> bool:
        return url.startswith("http://") or url.startswith("https://")

    def add_server(self, name: str, *, command: Optional[List[str]] = None,
                   url: Optional[str] = None) -> None:
        if command:
            self.servers[name] = {"type": "stdio", "command": command}
        elif url and self._valid_url(url):
            self.servers[name] = {"type": "http", "url": url}
        else:
            raise ValueError("need command=[...] or a valid http url")

    def list_servers(self) -> List[str]:
        return sorted(self.servers)

    # -- transport -----------------------------------------------------------

    def _next_id(self) -> int:
        with self._lock:
            self._id += 1
            return self._id

    def _call_stdio_service(self, name: str, method: str,
                            params: Dict[str, Any],
                            timeout: float = STDIO_TIMEOUT_S) -> Any:
        cfg = self.servers[name]
        proc = self.procs.start(name, cfg["command"])
        req_id = self._next_id()
        request = {"jsonrpc": "2.0", "id": req_id, "method": method,
                   "params": params}
        with self._lock:
            try:
                proc.stdin.write(json.dumps(request) + "\n")
                proc.stdin.flush()
            except (BrokenPipeError, OSError) as e:
                self.procs.stop(name)
                raise MCPError(f"stdio server {name} died: {e}")
            deadline = time.time() + timeout
            while time.time() < deadline:
                line = proc.stdout.readline()
                if not line:
                    if proc.poll() is not None:
                        raise MCPError(f"stdio server {name} exited")
                    time.sleep(0.05)
                    continue
                try:
                    msg = json.loads(line)
                except json.JSONDecodeError:
                    continue                      # skip log lines
                if msg.get("id") != req_id:
                    continue                      # stale response
                if "error" in msg:
                    raise MCPError(str(msg["error"]))
                return msg.get("result")
        raise MCPError(f"stdio call to {name}.{method} timed out")

    def _call_http_service(self, name: str, method: str,
                           params: Dict[str, Any],
                           timeout: float = 30.0) -> Any:
        cfg = self.servers[name]
        request = {"jsonrpc": "2.0", "id": self._next_id(), "method": method,
                   "params": params}
        try:
            r = requests.post(cfg["url"], json=request, timeout=timeout)
            msg = r.json()
        except (requests.RequestException, ValueError) as e:
            raise MCPError(f"http MCP server {name} failed: {e}")
        if "error" in msg:
            raise MCPError(str(msg["error"]))
        return msg.get("result")

    def call_service(self, name: str, method: str,
                     params: Optional[Dict[str, Any]] = None) -> Any:
        if name not in self.servers:
            raise MCPError(f"unknown MCP server: {name}")
        params = params or {}
        if self.servers[name]["type"] == "stdio":
            try:
                return self._call_stdio_service(name, method, params)
            except MCPError as e:
                # one restart on a dead subprocess (crash between calls,
                # OOM-killed, etc.) — the reference re-raised and lost the
                # turn; a fresh process usually recovers stateless servers
                if "died" not in str(e) and "exited" not in str(e):
                    raise
                logger.warning("restarting stdio MCP server %s after: %s",
                               name, e)
                self.procs.stop(name)
                return self._call_stdio_service(name, method, params)
        return self._call_http_service(name, method, params)

    def stop_server(self, name: str) -> bool:
        return self.procs.stop(name)


# -- service facades ---------------------------------------------------------

class MCPMemoryService:
    """Knowledge-graph ops (reference: mcp.py:753-864)."""

    def __init__(self, client: MCPClient, server: str = "memory"):
        self.client = client
        self.server = server

    def create_entities(self, entities: List[Dict[str, Any]]) -> Any:
        return self.client.call_service(self.server, "create_entities",
                                        {"entities": entities})

    def create_relations(self, relations: List[Dict[str, Any]]) -> Any:
        return self.client.call_service(self.server, "create_relations",
                                        {"relations": relations})

    def add_observations(self, observations: List[Dict[str, Any]]) -> Any:
        return self.client.call_service(self.server, "add_observations",
                                        {"observations": observations})

    def read_graph(self) -> Any:
        return self.client.call_service(self.server, "read_graph", {})

    def search_nodes(self, query: str) -> Any:
        return self.client.call_service(self.server, "search_nodes",
                                        {"query": query})

    def delete_entities(self, names: List[str]) -> Any:
        return self.client.call_service(self.server, "delete_entities",
                                        {"entityNames": names})


class MCPFetchService:
    def __init__(self, client: MCPClient, server: str = "fetch"):
        self.client = client
        self.server = server

    def fetch(self, url: str, max_length: int = 20000) -> Any:
        return self.client.call_service(self.server, "fetch",
                                        {"url": url, "max_length": max_length})


class MCPBraveSearchService:
    """Web search with a direct-REST fallback (reference: mcp.py:911-1010).
    The reference shipped a hardcoded fallback API key (cli.py:589) — a
    known defect we do NOT replicate: no key, no fallback."""

    def __init__(self, client: MCPClient, server: str = "brave_search"):
        self.client = client
        self.server = server

    def search(self, query: str, count: int = 5, offset: int = 0) -> Any:
        try:
            return self.client.call_service(self.server, "brave_web_search",
                                            {"query": query, "count": count,
                                             "offset": offset})
        except MCPError:
            return self._direct_search(query, count, offset)

    def _direct_search(self, query: str, count: int, offset: int = 0) -> Any:
        api_key = os.environ.get("BRAVE_API_KEY", "")
        if not api_key:
            raise MCPError("brave search unavailable: no MCP server and no "
                           "BRAVE_API_KEY")
        r = requests.get(
            "https://api.search.brave.com/res/v1/web/search",
            params={"q": query, "count": count, "offset": offset},
            headers={"X-Subscription-Token": api_key}, timeout=15)
        r.raise_for_status()
        data = r.json()
        return [{"title": w.get("title"), "url": w.get("url"),
                 "description": w.get("description")}
                for w in data.get("web", {}).get("results", [])[:count]]


class MCPGitHubService:
    def __init__(self, client: MCPClient, server: str = "github"):
        self.client = client
        self.server = server

    def search_repositories(self, query: str) -> Any:
        return self.client.call_service(self.server, "search_repositories",
                                        {"query": query})

    def get_file_contents(self, owner: str, repo: str, path: str) -> Any:
        return self.client.call_service(self.server, "get_file_contents",
                                        {"owner": owner, "repo": repo,
                                         "path": path})


class MCPManager:
    """Facade bundle + registry wiring (reference: mcp.py:1097-1185)."""

    def __init__(self, client: Optional[MCPClient] = None):
        self.client = client or MCPClient()
        self.memory = MCPMemoryService(self.client)
        self.fetch = MCPFetchService(self.client)
        self.brave_search = MCPBraveSearchService(self.client)
        self.github = MCPGitHubService(self.client)

    def attach_registry(self, registry) -> None:
        """Route ``mcp_<server>_<method>`` tool names through the client
        (replaces the reference's hard-coded registry special-casing,
        registry.py:340-467)."""

        def hook(tool_name: str, args: Dict[str, Any]) -> Any:
            rest = tool_name[len("mcp_"):]
            for server in self.client.list_servers():
                if rest.startswith(server + "_"):
                    method = rest[len(server) + 1:]
                    return {"result": self.client.call_service(server, method, args)}
            raise MCPError(f"no MCP server matches tool {tool_name}")

        registry.register_prefix_hook("mcp_", hook)
        # the reference special-cases the bare "brave_web_search" tool name
        # (registry.py:340-467); same surface via a prefix hook:
        registry.register_prefix_hook(
            "brave_web_search",
            lambda name, args: {"results": self.brave_search.search(
                args.get("query", ""), count=int(args.get("count", 5)),
                offset=int(args.get("offset", 0)))})

    def shutdown(self) -> None:
        self.client.procs.stop_all()
