"""Thin handlers: unpack tool args -> call the singleton implementations.

Parity: fei/tools/handlers.py:49-590. Notable: smart_search builds
language-specific definition regexes from a natural query (handlers.py:308-417)
and repo_deps aggregates module-level imports (handlers.py:469-540).
"""

from __future__ import annotations

import os
import re
from typing import Any, Dict, List

from fei_amd.tools import code as c
from fei_amd.tools import repomap


def glob_tool_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    files = c.glob_finder.find(args["pattern"], args.get("path"))
    return {"pattern": args["pattern"], "count": len(files), "files": files[:500]}


def grep_tool_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    matches = c.grep_tool.search(args["pattern"], args.get("path"), args.get("include"))
    return {"pattern": args["pattern"], "count": len(matches), "matches": matches[:500]}


def view_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.file_viewer.view(args["file_path"], args.get("offset", 1), args.get("limit", 2000))


def edit_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.code_editor.edit_file(args["file_path"], args["old_string"], args["new_string"])


def replace_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.code_editor.replace_file(args["file_path"], args["content"])


def ls_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.directory_explorer.list_directory(args["path"], args.get("ignore"))


def regex_edit_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.code_editor.regex_replace(
        args["file_path"], args["pattern"], args["replacement"],
        count=args.get("count", 0), validate=args.get("validate", True),
        validators=args.get("validators"),
    )


def batch_glob_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    results = c.glob_finder.batch(args["patterns"], args.get("path"))
    limit = int(args.get("limit_per_pattern", 20))   # reference default
    return {"results": {k: v[:limit] for k, v in results.items()}}


def find_in_files_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return {"results": c.grep_tool.find_in_files(
        args["files"], args["pattern"],
        case_sensitive=bool(args.get("case_sensitive", False)))}


# -- SmartSearch -------------------------------------------------------------

_DEF_PATTERNS = {
    "python": [r"def\s+{q}\w*\s*\(", r"class\s+{q}\w*\s*[(:]", r"^{q}\w*\s*="],
    "js": [r"function\s+{q}\w*\s*\(", r"class\s+{q}\w*", r"(?:const|let|var)\s+{q}\w*\s*="],
    "c": [r"\w+\s+{q}\w*\s*\([^;]*\)\s*\{{", r"#define\s+{q}\w*", r"(?:struct|enum|union)\s+{q}\w*"],
    "go": [r"func\s+(?:\(\w+ \*?\w+\)\s*)?{q}\w*\s*\(", r"type\s+{q}\w*"],
    "rust": [r"fn\s+{q}\w*\s*[(<]", r"(?:struct|enum|trait|impl)\s+{q}\w*"],
}
_LANG_EXT = {
    "python": ("*.py",), "js": ("*.js", "*.ts", "*.jsx", "*.tsx"),
    "c": ("*.c", "*.h", "*.cpp", "*.hpp", "*.cc", "*.hip", "*.cu"),
    "go": ("*.go",), "rust": ("*.rs",),
}


def smart_search_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    query = args["query"].strip()
    path = args.get("path") or os.getcwd()
    lang = args.get("language")
    # Pull the identifier-ish core of the query ("def parse_args" -> parse_args).
    words = re.findall(r"[A-Za-z_][A-Za-z0-9_]*", query)
    keywords = [w for w in words if w not in
                ("def", "class", "function", "fn", "func", "the", "a", "an", "find", "where", "is")]
    ident = keywords[-1] if keywords else query
    langs = [lang] if lang in _DEF_PATTERNS else list(_DEF_PATTERNS)
    definitions: List[Dict[str, Any]] = []
    for lg in langs:
        pattern = "|".join(p.format(q=re.escape(ident)) for p in _DEF_PATTERNS[lg])
        for ext in _LANG_EXT[lg]:
            for m in c.grep_tool.search(pattern, path, include=ext)[:50]:
                m["language"] = lg
                definitions.append(m)
    plain = c.grep_tool.search(re.escape(ident), path)[:50] if ident else []
    ctx = (args.get("context") or "").strip().lower()
    if ctx:
        ctx_words = set(re.findall(r"[a-z_][a-z0-9_]*", ctx))

        def ctx_rank(m):
            hay = (m.get("file", "") + " " + m.get("line", "")).lower()
            return -sum(1 for w in ctx_words if w in hay)

        definitions.sort(key=ctx_rank)
        plain.sort(key=ctx_rank)
    return {
        "query": query, "identifier": ident,
        "definitions": definitions[:100], "mentions": plain,
    }


def repo_map_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    mapper = repomap.RepoMapper(args.get("path") or os.getcwd(),
                                exclude_patterns=args.get("exclude_patterns"))
    budget = args.get("token_budget", 2000)
    return {"map": mapper.generate_map(token_budget=budget)}


def repo_summary_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    out = repomap.RepoMapper(
        args.get("path") or os.getcwd(),
        exclude_patterns=args.get("exclude_patterns")).summary()
    # max_tokens (reference field): truncate the largest-files list as a
    # crude size control (the summary is already compact)
    if args.get("max_tokens"):
        out["largest_files"] = out["largest_files"][:5]
    return out


def repo_deps_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    mapper = repomap.RepoMapper(args.get("path") or os.getcwd())
    deps = mapper.dependencies()
    module = args.get("module")
    if module:
        deps = {k: v for k, v in deps.items() if module in k}
    depth = int(args.get("depth", 1))
    if depth > 1 and module:
        # expand transitively: pull in files whose module names appear in
        # the already-selected dependency lists
        for _ in range(depth - 1):
            wanted = {m for vs in deps.values() for m in vs}
            for k, v in mapper.dependencies().items():
                stem = os.path.splitext(os.path.basename(k))[0]
                if stem in wanted and k not in deps:
                    deps[k] = v
    return {"dependencies": deps}


def shell_handler(args: Dict[str, Any]) -> Dict[str, Any]:
    return c.shell_runner.run(
        args["command"],
        timeout=float(args.get("timeout", 60.0)),
        background=bool(args.get("background", False)),
        working_dir=args.get("working_dir") or args.get("current_dir"),
    )
