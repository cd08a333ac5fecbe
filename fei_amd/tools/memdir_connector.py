"""HTTP client for the memdir server + server subprocess lifecycle.

Parity: reference MemdirConnector (fei/tools/memdir_connector.py:25-620):
class-level server-process singleton, X-API-Key requests, port probe,
detached ``python -m fei_amd.memdir.run_server`` spawn with health polling,
atexit stop, REST wrappers, and the start/stop/status commands.
"""

from __future__ import annotations

import atexit
import os
import socket
import subprocess
import sys
import threading
import time
from typing import Any, Dict, Optional

import requests

from fei_amd.utils.config import get_config
from fei_amd.utils.logging import get_logger

logger = get_logger("tools.memdir_connector")


class MemdirConnector:
    _server_process: Optional[subprocess.Popen] = None
    _server_lock = threading.Lock()

    def __init__(self, server_url: Optional[str] = None,
                 api_key: Optional[str] = None,
                 base: Optional[str] = None,
                 auto_start: bool = False):
        cfg = get_config()
        port = cfg.get_typed("memdir.server_port", 5000)
        self.server_url = (server_url or
                           os.environ.get("MEMDIR_SERVER_URL") or
                           f"http://127.0.0.1:{port}")
        self.api_key = api_key if api_key is not None else \
            (os.environ.get("MEMDIR_API_KEY") or cfg.get("memdir.api_key", ""))
        self.base = base
        self.auto_start = auto_start

    # -- plumbing ------------------------------------------------------------

    def _headers(self) -> Dict[str, str]:
        return {"X-API-Key": self.api_key} if self.api_key else {}

    def _make_request(self, method: str, path: str, timeout: float = 10.0,
                      **kwargs) -> Dict[str, Any]:
        if self.auto_start:
            self.start_server_command()
        url = self.server_url.rstrip("/") + path
        try:
            r = requests.request(method, url, headers=self._headers(),
                                 timeout=timeout, **kwargs)
        except requests.RequestException as e:
            return {"error": f"memdir server unreachable: {e}"}
        try:
            data = r.json()
        except ValueError:
            data = {"error": f"non-JSON response (status {r.status_code})"}
        if not r.ok and "error" not in data:
            data["error"] = f"HTTP {r.status_code}"
        return data

    def _port(self) -> int:
        return int(self.server_url.rsplit(":", 1)[-1].rstrip("/"))

    def _port_in_use(self) -> bool:
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
            s.settimeout(0.25)
            return s.connect_ex(("127.0.0.1", self._port())) == 0

    # -- server lifecycle ----------------------------------------------------

    def start_server_command(self, wait_s: float = 5.0) -> Dict[str, Any]:
        """Spawn the memdir server detached if its port is free; health-poll
        (reference: memdir_connector.py:141-239)."""
        with MemdirConnector._server_lock:
            if self._port_in_use():
                return {"success": True, "message": "server already running"}
            cmd = [sys.executable, "-m", "fei_amd.memdir.run_server",
                   "--port", str(self._port())]
            if self.api_key:
                cmd += ["--api-key", self.api_key]
            if self.base:
                cmd += ["--base", self.base]
            proc = subprocess.Popen(cmd, stdout=subprocess.DEVNULL,
                                    stderr=subprocess.DEVNULL,
                                    start_new_session=True)
            MemdirConnector._server_process = proc
            atexit.register(MemdirConnector._stop_server)
        deadline = time.time() + wait_s
        while time.time() < deadline:
            try:
                r = requests.get(self.server_url + "/health", timeout=0.5)
                if r.ok:
                    return {"success": True, "message": "server started",
                            "pid": proc.pid}
            except requests.RequestException:
                pass
            time.sleep(0.25)
        return {"error": "server did not become healthy in time"}

    @classmethod
    def _stop_server(cls) -> None:
        with cls._server_lock:
            proc = cls._server_process
            cls._server_process = None
        if proc and proc.poll() is None:
            proc.terminate()
            try:
                proc.wait(timeout=3)
            except subprocess.TimeoutExpired:
                proc.kill()

    def stop_server_command(self) -> Dict[str, Any]:
        MemdirConnector._stop_server()
        return {"success": True, "message": "server stopped"}

    def get_server_status(self) -> Dict[str, Any]:
        if not self._port_in_use():
            return {"running": False}
        health = self._make_request("GET", "/health")
        return {"running": "error" not in health, **health}

    # -- REST wrappers -------------------------------------------------------

    def check_connection(self) -> bool:
        return "error" not in self._make_request("GET", "/health", timeout=2.0)

    def list_memories(self, folder: str = "", status: str = "cur",
                      with_content: bool = False) -> Dict[str, Any]:
        return self._make_request("GET", "/memories", params={
            "folder": folder, "status": status,
            "with_content": int(with_content)})

    def create_memory(self, headers: Dict[str, str], body: str,
                      folder: str = "", flags: str = "") -> Dict[str, Any]:
        return self._make_request("POST", "/memories", json={
            "headers": headers, "body": body, "folder": folder,
            "flags": flags})

    def get_memory(self, memory_id: str) -> Dict[str, Any]:
        return self._make_request("GET", f"/memories/{memory_id}")

    def move_memory(self, memory_id: str, folder: str) -> Dict[str, Any]:
        return self._make_request("PUT", f"/memories/{memory_id}",
                                  json={"folder": folder})

    def update_flags(self, memory_id: str, flags: str) -> Dict[str, Any]:
        return self._make_request("PUT", f"/memories/{memory_id}",
                                  json={"flags": flags})

    def delete_memory(self, memory_id: str,
                      permanent: bool = False) -> Dict[str, Any]:
        return self._make_request("DELETE", f"/memories/{memory_id}",
                                  params={"permanent": int(permanent)})

    def search(self, query: str, folder: Optional[str] = None,
               with_content: bool = False) -> Dict[str, Any]:
        params: Dict[str, Any] = {"q": query}
        if folder:
            params["folder"] = folder
        if with_content:
            params["with_content"] = 1
        return self._make_request("GET", "/search", params=params)

    def folders(self) -> Dict[str, Any]:
        return self._make_request("GET", "/folders")

    def create_folder(self, name: str) -> Dict[str, Any]:
        return self._make_request("POST", "/folders", json={"name": name})

    def folder_stats(self, name: str) -> Dict[str, Any]:
        return self._make_request("GET", f"/folders/{name}/stats")

    def run_filters(self) -> Dict[str, Any]:
        return self._make_request("POST", "/filters/run")
