"""memdir HTTP server: Flask REST facade over the memdir primitives.

Route parity with the reference (memdir_tools/server.py:67-370):
  GET  /health
  GET  /memories?folder=&status=&with_content=
  POST /memories                  {headers, body|content, folder, flags}
  GET  /memories/<id>
  PUT  /memories/<id>             {folder} (move) or {flags} (reflag)
  DELETE /memories/<id>           (move to .Trash; ?permanent=1 to unlink)
  GET  /search?q=<query language>
  GET  /folders ; POST /folders ; DELETE /folders/<name> ; GET /folders/<name>/stats
  POST /filters/run

Auth: ``X-API-Key`` header checked with hmac.compare_digest (the reference
imported the long-removed werkzeug ``safe_str_cmp`` — a known defect we fix;
SURVEY.md defect list / server.py:14).
"""

from __future__ import annotations

import hmac
import os
from typing import Any, Dict, Optional

from flask import Flask, jsonify, request

from fei_amd.memdir import utils as mu
from fei_amd.memdir import search as msearch
from fei_amd.memdir.folders import MemdirFolderManager
from fei_amd.memdir.filter import run_filters
from fei_amd.utils.logging import get_logger

logger = get_logger("memdir.server")


def create_app(base: Optional[str] = None, api_key: Optional[str] = None) -> Flask:
    app = Flask("fei_amd.memdir")
    key = api_key if api_key is not None else os.environ.get("MEMDIR_API_KEY", "")
    mgr = MemdirFolderManager(base)

    def authorized() -> bool:
        if not key:
            return True
        provided = request.headers.get("X-API-Key", "")
        return hmac.compare_digest(provided, key)

    @app.errorhandler(ValueError)
    def _bad_value(e):
        # traversal-rejecting folder/status/filename validation
        # (fei_amd.memdir.utils.folder_path) surfaces as a 400, not a 500
        return jsonify({"error": str(e)}), 400

    @app.before_request
    def _check_key():
        if request.path == "/health":
            return None
        if not authorized():
            return jsonify({"error": "unauthorized"}), 401
        return None

    # -- health --------------------------------------------------------------

    @app.get("/health")
    def health():
        return jsonify({"status": "ok", "base": mu.get_memdir_base(base)})

    # -- memories ------------------------------------------------------------

    @app.get("/memories")
    def list_memories_route():
        folder = request.args.get("folder", "")
        status = request.args.get("status", "cur")
        with_content = request.args.get("with_content", "0") in ("1", "true")
        mems = mu.list_memories(folder, status, include_content=with_content, base=base)
        return jsonify({"count": len(mems), "memories": mems})

    @app.post("/memories")
    def create_memory_route():
        data: Dict[str, Any] = request.get_json(force=True, silent=True) or {}
        headers = data.get("headers", {})
        body = data.get("body", data.get("content", ""))
        folder = data.get("folder", "")
        flags = data.get("flags", "")
        filename = mu.create_memory(folder, headers, body, flags, base=base)
        return jsonify({"success": True, "filename": filename, "folder": folder}), 201

    @app.get("/memories/<memory_id>")
    def get_memory_route(memory_id: str):
        loc = mu.find_memory(memory_id, base=base)
        if loc is None:
            return jsonify({"error": "not found"}), 404
        mem = mu.read_memory(*loc, base=base)
        return jsonify(mem)

    @app.put("/memories/<memory_id>")
    def update_memory_route(memory_id: str):
        data: Dict[str, Any] = request.get_json(force=True, silent=True) or {}
        loc = mu.find_memory(memory_id, base=base)
        if loc is None:
            return jsonify({"error": "not found"}), 404
        folder, status, filename = loc
        if "folder" in data:
            ok = mu.move_memory(filename, folder, data["folder"], src_status=status,
                                dst_status=data.get("status", "cur"), base=base)
            return jsonify({"success": ok, "folder": data["folder"]})
        if "flags" in data:
            new_name = mu.update_memory_flags(filename, folder, status, data["flags"], base=base)
            return jsonify({"success": new_name is not None, "filename": new_name})
        return jsonify({"error": "nothing to update (need folder or flags)"}), 400

    @app.delete("/memories/<memory_id>")
    def delete_memory_route(memory_id: str):
        loc = mu.find_memory(memory_id, base=base)
        if loc is None:
            return jsonify({"error": "not found"}), 404
        folder, status, filename = loc
        if request.args.get("permanent", "0") in ("1", "true"):
            path = os.path.join(mu.folder_path(folder, base), status,
                                filename)
            os.unlink(path)
            return jsonify({"success": True, "permanent": True})
        ok = mu.move_memory(filename, folder, ".Trash", src_status=status,
                            dst_status="cur", base=base)
        return jsonify({"success": ok, "folder": ".Trash"})

    # -- search --------------------------------------------------------------

    @app.get("/search")
    def search_route():
        q = request.args.get("q", "")
        query = msearch.parse_search_args(q)
        if request.args.get("folder"):
            query.folders = [request.args["folder"]]
        if request.args.get("with_content", "0") in ("1", "true"):
            query.with_content = True
        results = msearch.search_memories(query, base=base)
        return jsonify({"count": len(results), "results": results})

    # -- folders -------------------------------------------------------------

    @app.get("/folders")
    def folders_route():
        return jsonify({"folders": mgr.list_folders()})

    @app.post("/folders")
    def create_folder_route():
        data = request.get_json(force=True, silent=True) or {}
        name = data.get("name", "")
        if not name:
            return jsonify({"error": "missing folder name"}), 400
        ok = mgr.create_folder(name)
        return jsonify({"success": ok, "folder": name}), 201 if ok else 400

    @app.delete("/folders/<path:name>")
    def delete_folder_route(name: str):
        ok = mgr.delete_folder(name)
        return jsonify({"success": ok})

    @app.get("/folders/<path:name>/stats")
    def folder_stats_route(name: str):
        if name == "root":
            name = ""
        return jsonify(mgr.get_folder_stats(name))

    # -- filters -------------------------------------------------------------

    @app.post("/filters/run")
    def run_filters_route():
        return jsonify(run_filters(base=base))

    return app
