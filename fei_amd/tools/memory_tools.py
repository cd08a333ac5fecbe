"""Memory tools exposed to the model + the MemoryManager facade.

Parity: reference memory_tools.py registers 8 memory tools
(memory_search/create/view/list/delete/search_by_tag and the server
start/stop/status commands, fei/tools/memory_tools.py:23-610) and a
MemoryManager dual memdir+memorychain facade with save_conversation
(:613-812).

Design difference: tool handlers access the memdir tree DIRECTLY (same
process, no HTTP hop) — the reference spawned a Flask server per handler
call path (memory_tools.py:157-165), which we keep only for the explicit
server commands. ``memory_semantic_search`` is new: top-k over the GPU
embedding index (fei_amd/memdir/embed_index.py).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from fei_amd.memdir import search as msearch
from fei_amd.memdir import utils as mu
from fei_amd.utils.logging import get_logger

logger = get_logger("tools.memory_tools")

MEMORY_SEARCH_TOOL = {
    "name": "memory_search",
    "description": ("Search stored memories with the memdir query language "
                    "(keywords, #tag, +F flags, field:value, /regex/, "
                    "sort:, limit:)."),
    "input_schema": {
        "type": "object",
        "properties": {
            "query": {"type": "string", "description": "Query string"},
            "folder": {"type": "string", "description": "Restrict to one folder"},
            "status": {"type": "string", "enum": ["cur", "new", "tmp"]},
            "limit": {"type": "integer", "description": "Max results (default 50)"},
            "with_content": {"type": "boolean"},
        },
        "required": ["query"],
    },
}

MEMORY_SEMANTIC_SEARCH_TOOL = {
    "name": "memory_semantic_search",
    "description": ("Semantic search over memories via the GPU embedding "
                    "index (cosine top-k). Build the index first with "
                    "memory_index_build."),
    "input_schema": {
        "type": "object",
        "properties": {
            "query": {"type": "string"},
            "topk": {"type": "integer"},
        },
        "required": ["query"],
    },
}

MEMORY_INDEX_BUILD_TOOL = {
    "name": "memory_index_build",
    "description": ("(Re)build the search indexes over all memories: the "
                    "semantic embedding index and the lexical FTS index."),
    "input_schema": {"type": "object", "properties": {}},
}

MEMORY_KEYWORD_SEARCH_TOOL = {
    "name": "memory_keyword_search",
    "description": ("Ranked keyword search (bm25 over subject/tags/content "
                    "via the FTS index; one query instead of a corpus "
                    "scan). Build with memory_index_build."),
    "input_schema": {
        "type": "object",
        "properties": {
            "query": {"type": "string"},
            "limit": {"type": "integer"},
            "with_content": {"type": "boolean"},
        },
        "required": ["query"],
    },
}

MEMORY_CREATE_TOOL = {
    "name": "memory_create",
    "description": "Store a new memory with subject, tags and body text.",
    "input_schema": {
        "type": "object",
        "properties": {
            "subject": {"type": "string"},
            "body": {"type": "string"},
            "content": {"type": "string", "description": "Alias of body (reference field name)"},
            "tags": {"type": "string", "description": "comma-separated"},
            "folder": {"type": "string"},
            "priority": {"type": "string", "enum": ["low", "normal", "high"]},
            "flags": {"type": "string", "description": "subset of SRFP"},
        },
        "required": ["subject"],
    },
}

MEMORY_VIEW_TOOL = {
    "name": "memory_view",
    "description": "View one memory by its id (8-hex unique or full filename).",
    "input_schema": {
        "type": "object",
        "properties": {
            "memory_id": {"type": "string"},
            "folder": {"type": "string", "description": "Restrict the lookup to one folder"},
        },
        "required": ["memory_id"],
    },
}

MEMORY_LIST_TOOL = {
    "name": "memory_list",
    "description": "List memories in a folder (default root, status cur+new).",
    "input_schema": {
        "type": "object",
        "properties": {
            "folder": {"type": "string"},
            "status": {"type": "string", "enum": ["cur", "new", "tmp"]},
            "limit": {"type": "integer", "description": "Max results (default 50)"},
        },
    },
}

MEMORY_DELETE_TOOL = {
    "name": "memory_delete",
    "description": "Move a memory to .Trash (or delete permanently).",
    "input_schema": {
        "type": "object",
        "properties": {
            "memory_id": {"type": "string"},
            "folder": {"type": "string", "description": "Restrict the lookup to one folder"},
            "permanent": {"type": "boolean"},
        },
        "required": ["memory_id"],
    },
}

MEMORY_SEARCH_BY_TAG_TOOL = {
    "name": "memory_search_by_tag",
    "description": "List memories carrying a tag.",
    "input_schema": {
        "type": "object",
        "properties": {"tag": {"type": "string"}},
        "required": ["tag"],
    },
}

SERVER_TOOLS = [
    {"name": "memdir_server_start", "description": "Start the memdir HTTP server.",
     "input_schema": {"type": "object", "properties": {}}},
    {"name": "memdir_server_stop", "description": "Stop the memdir HTTP server.",
     "input_schema": {"type": "object", "properties": {}}},
    {"name": "memdir_server_status", "description": "Memdir HTTP server status.",
     "input_schema": {"type": "object", "properties": {}}},
]


class MemoryTools:
    """Direct-FS handlers bound to one memdir base."""

    def __init__(self, base: Optional[str] = None):
        self.base = base
        self._index = None
        self._fts = None

    # lazily built embedding index
    def index(self):
        if self._index is None:
            from fei_amd.memdir.embed_index import EmbeddingIndex
            self._index = EmbeddingIndex(base=self.base)
        return self._index

    def search(self, args: Dict[str, Any]) -> Dict[str, Any]:
        q = msearch.parse_search_args(args["query"])
        if args.get("folder"):
            q.folders = [args["folder"]]
        if args.get("status"):
            q.statuses = [args["status"]]
        if args.get("with_content"):
            q.with_content = True
        results = msearch.search_memories(q, base=self.base)
        limit = int(args.get("limit", 50))
        return {"count": len(results), "results": results[:limit]}

    def semantic_search(self, args: Dict[str, Any]) -> Dict[str, Any]:
        results = self.index().search_memories(args["query"],
                                               topk=args.get("topk", 10))
        if not results and self.index().embeddings is None:
            return {"error": "no semantic index; run memory_index_build first"}
        return {"count": len(results), "results": results}

    def fts(self):
        if self._fts is None:
            from fei_amd.memdir.fts_index import FtsIndex
            self._fts = FtsIndex(base=self.base)
        return self._fts

    def index_build(self, args: Dict[str, Any]) -> Dict[str, Any]:
        n = self.index().build()
        n_fts = self.fts().build()
        return {"success": True, "indexed": n, "fts_indexed": n_fts}

    def keyword_search(self, args: Dict[str, Any]) -> Dict[str, Any]:
        res = self.fts().search_memories(
            args["query"], limit=int(args.get("limit", 20)),
            with_content=bool(args.get("with_content", True)))
        return {"count": len(res), "results": res}

    def create(self, args: Dict[str, Any]) -> Dict[str, Any]:
        headers = {"Subject": args["subject"]}
        if args.get("tags"):
            headers["Tags"] = args["tags"]
        if args.get("priority"):
            headers["Priority"] = args["priority"]
        body = args.get("body") or args.get("content", "")
        filename = mu.create_memory(args.get("folder", ""), headers,
                                    body, flags=args.get("flags", ""),
                                    base=self.base)
        meta = mu.parse_memory_filename(filename)
        # keep a live semantic index fresh (full rebuilds stay explicit
        # via memory_index_build; delete() drops rows symmetrically)
        if self._index is not None and self._index.embeddings is not None:
            text = (headers.get("Subject", "") + "\n" +
                    headers.get("Tags", "") + "\n" + body)[:2000]
            folder = args.get("folder", "")
            self._index.add_texts(
                [text], [f"{folder}\x00new\x00{filename}"])
        if self._fts is not None and self._fts.count() > 0:
            self._fts.add(args.get("folder", ""), "new", filename,
                          headers.get("Subject", ""),
                          headers.get("Tags", ""), body)
        return {"success": True, "memory_id": meta["unique"],
                "filename": filename}

    def view(self, args: Dict[str, Any]) -> Dict[str, Any]:
        loc = mu.find_memory(args["memory_id"], base=self.base,
                             folder=args.get("folder"))
        if loc is None:
            return {"error": f"memory not found: {args['memory_id']}"}
        return mu.read_memory(*loc, base=self.base)

    def list(self, args: Dict[str, Any]) -> Dict[str, Any]:
        folder = args.get("folder", "")
        status = args.get("status")
        statuses = [status] if status else ["cur", "new"]
        out: List[Dict[str, Any]] = []
        for st in statuses:
            out.extend(mu.list_memories(folder, st, include_content=True,
                                        base=self.base))
        out = out[: int(args.get("limit", 50))]
        brief = [{"memory_id": m["metadata"]["unique"],
                  "subject": m.get("headers", {}).get("Subject", ""),
                  "tags": m.get("headers", {}).get("Tags", ""),
                  "status": m["status"],
                  "flags": "".join(m["metadata"]["flags"])} for m in out]
        return {"folder": folder or "(root)", "count": len(brief),
                "memories": brief}

    def delete(self, args: Dict[str, Any]) -> Dict[str, Any]:
        loc = mu.find_memory(args["memory_id"], base=self.base,
                             folder=args.get("folder"))
        if loc is None:
            return {"error": f"memory not found: {args['memory_id']}"}
        folder, status, filename = loc
        if self._index is not None and self._index.embeddings is not None:
            # maildir moves keep the unique part of the filename stable
            self._index.remove(filename.split(":", 1)[0])
        if self._fts is not None and self._fts.count() > 0:
            self._fts.remove(filename.split(":", 1)[0])
        if args.get("permanent"):
            import os
            root = mu.get_memdir_base(self.base)
            os.unlink(os.path.join(root, folder, status, filename)
                      if folder else os.path.join(root, status, filename))
            return {"success": True, "permanent": True}
        ok = mu.move_memory(filename, folder, ".Trash", src_status=status,
                            dst_status="cur", base=self.base)
        return {"success": ok, "folder": ".Trash"}

    def search_by_tag(self, args: Dict[str, Any]) -> Dict[str, Any]:
        results = msearch.search(f"#{args['tag']}", base=self.base)
        return {"count": len(results), "results": results[:50]}


def create_memory_tools(registry, base: Optional[str] = None,
                        connector=None) -> MemoryTools:
    """Register the memory tools (reference: memory_tools.py:526-610)."""
    tools = MemoryTools(base=base)
    pairs = [
        (MEMORY_SEARCH_TOOL, tools.search),
        (MEMORY_SEMANTIC_SEARCH_TOOL, tools.semantic_search),
        (MEMORY_KEYWORD_SEARCH_TOOL, tools.keyword_search),
        (MEMORY_INDEX_BUILD_TOOL, tools.index_build),
        (MEMORY_CREATE_TOOL, tools.create),
        (MEMORY_VIEW_TOOL, tools.view),
        (MEMORY_LIST_TOOL, tools.list),
        (MEMORY_DELETE_TOOL, tools.delete),
        (MEMORY_SEARCH_BY_TAG_TOOL, tools.search_by_tag),
    ]
    for spec, handler in pairs:
        registry.register_tool(spec["name"], spec["description"],
                               spec["input_schema"], handler)

    from fei_amd.tools.memdir_connector import MemdirConnector
    conn = connector or MemdirConnector(base=base)
    registry.register_tool(SERVER_TOOLS[0]["name"], SERVER_TOOLS[0]["description"],
                           SERVER_TOOLS[0]["input_schema"],
                           lambda a: conn.start_server_command())
    registry.register_tool(SERVER_TOOLS[1]["name"], SERVER_TOOLS[1]["description"],
                           SERVER_TOOLS[1]["input_schema"],
                           lambda a: conn.stop_server_command())
    registry.register_tool(SERVER_TOOLS[2]["name"], SERVER_TOOLS[2]["description"],
                           SERVER_TOOLS[2]["input_schema"],
                           lambda a: conn.get_server_status())
    return tools


class MemoryManager:
    """Dual memdir + memorychain facade (reference: memory_tools.py:613-812)."""

    def __init__(self, base: Optional[str] = None, chain_connector=None):
        self.tools = MemoryTools(base=base)
        self.base = base
        self.chain = chain_connector

    def save_memory(self, subject: str, body: str = "", tags: str = "",
                    folder: str = "", to_chain: bool = False) -> Dict[str, Any]:
        out = self.tools.create({"subject": subject, "body": body,
                                 "tags": tags, "folder": folder})
        if to_chain and self.chain is not None and self.chain.is_available():
            out["chain"] = self.chain.add_memory(
                {"Subject": subject, "Tags": tags}, body=body)
        return out

    def save_conversation(self, messages: List[Dict[str, Any]],
                          subject: Optional[str] = None,
                          tags: str = "conversation") -> Dict[str, Any]:
        """Persist a conversation transcript as one memory
        (reference: memory_tools.py:749-781)."""
        subject = subject or f"Conversation {time.strftime('%Y-%m-%d %H:%M')}"
        lines = []
        for msg in messages:
            role = msg.get("role", "?")
            content = msg.get("content", "")
            if isinstance(content, list):
                parts = []
                for b in content:
                    if isinstance(b, dict):
                        if b.get("type") == "text":
                            parts.append(str(b.get("text", "")))
                        elif b.get("type") == "tool_use":
                            parts.append(f"[tool_use {b.get('name')}]")
                        elif b.get("type") == "tool_result":
                            parts.append("[tool_result]")
                content = " ".join(parts)
            lines.append(f"{role}: {str(content)[:1000]}")
        return self.save_memory(subject, body="\n".join(lines), tags=tags)

    def recall(self, query: str, semantic: bool = False,
               topk: int = 10) -> List[Dict[str, Any]]:
        if semantic:
            return self.tools.semantic_search({"query": query, "topk": topk}) \
                .get("results", [])
        return self.tools.search({"query": query, "with_content": True}) \
            .get("results", [])
