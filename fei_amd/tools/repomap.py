"""Repository map under a token budget.

Parity: the reference's aider-style RepoMapper (fei/tools/repomap.py:68-544):
file walk with excludes, symbol extraction, a cross-file symbol-reference
dependency graph, PageRank-ish file ranking, and budgeted formatting.

Symbol extraction mirrors the reference's ladder (repomap.py:160-281,
324-389): Python via the exact ``ast`` parser, every other language via
tree-sitter per-language queries WHEN ``tree_sitter_languages`` is
importable (optional in this image), regex table as the final fallback.
"""

from __future__ import annotations

import ast
import os
import re
from collections import defaultdict
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, List, Optional, Set, Tuple

from fei_amd.utils.logging import get_logger

logger = get_logger("tools.repomap")

EXCLUDE_DIRS = {".git", "__pycache__", "node_modules", ".venv", "venv", "build",
                "dist", ".pytest_cache", ".fei_backups", "gpurun_out"}
SOURCE_EXTS = {".py", ".js", ".ts", ".jsx", ".tsx", ".c", ".h", ".cpp", ".hpp",
               ".cc", ".hip", ".cu", ".go", ".rs", ".java", ".rb", ".sh"}

_REGEX_SYMBOLS = [
    (re.compile(r"^\s*def\s+([A-Za-z_]\w*)\s*\(", re.M), "function"),
    (re.compile(r"^\s*class\s+([A-Za-z_]\w*)", re.M), "class"),
    (re.compile(r"^\s*function\s+([A-Za-z_$]\w*)\s*\(", re.M), "function"),
    (re.compile(r"^\s*(?:export\s+)?(?:const|let|var)\s+([A-Za-z_$]\w*)\s*=\s*(?:async\s*)?\(", re.M), "function"),
    (re.compile(r"^\s*fn\s+([A-Za-z_]\w*)", re.M), "function"),
    (re.compile(r"^\s*func\s+(?:\([^)]*\)\s*)?([A-Za-z_]\w*)\s*\(", re.M), "function"),
    (re.compile(r"^\s*(?:pub\s+)?(?:struct|enum|trait)\s+([A-Za-z_]\w*)", re.M), "class"),
    (re.compile(r"^__global__\s+\w+\s+([A-Za-z_]\w*)\s*\(", re.M), "kernel"),
]

_IMPORT_RE = [
    re.compile(r"^\s*import\s+([\w.]+)", re.M),
    re.compile(r"^\s*from\s+([\w.]+)\s+import", re.M),
    re.compile(r"""^\s*(?:import|export).*?from\s+['"]([^'"]+)['"]""", re.M),
    re.compile(r"""^\s*#include\s+[<"]([^>"]+)[>"]""", re.M),
]

# -- tree-sitter symbol extraction (reference repomap.py:160-281) ------------
# Used when ``tree_sitter_languages`` is importable (it is optional in this
# image); per-language capture queries mirror the reference's set
# (repomap.py:244-281). Python keeps the exact ``ast`` extractor; everything
# else prefers tree-sitter and falls back to the regex table.

_TS_EXT_LANG = {
    ".js": "javascript", ".jsx": "javascript",
    ".ts": "typescript", ".tsx": "tsx",
    ".c": "c", ".h": "c",
    ".cpp": "cpp", ".hpp": "cpp", ".cc": "cpp",
    ".go": "go", ".rs": "rust", ".java": "java", ".rb": "ruby",
}

_TS_QUERIES = {
    "javascript": """
        (function_declaration name: (identifier) @function)
        (generator_function_declaration name: (identifier) @function)
        (class_declaration name: (identifier) @class)
        (method_definition name: (property_identifier) @method)
        (variable_declarator name: (identifier) @function
                             value: (arrow_function))
    """,
    "typescript": """
        (function_declaration name: (identifier) @function)
        (class_declaration name: (type_identifier) @class)
        (interface_declaration name: (type_identifier) @class)
        (method_definition name: (property_identifier) @method)
    """,
    "tsx": """
        (function_declaration name: (identifier) @function)
        (class_declaration name: (type_identifier) @class)
    """,
    "c": """
        (function_definition declarator:
            (function_declarator declarator: (identifier) @function))
        (struct_specifier name: (type_identifier) @class)
        (enum_specifier name: (type_identifier) @class)
    """,
    "cpp": """
        (function_definition declarator:
            (function_declarator declarator: (identifier) @function))
        (function_definition declarator:
            (function_declarator declarator:
                (qualified_identifier) @method))
        (class_specifier name: (type_identifier) @class)
        (struct_specifier name: (type_identifier) @class)
    """,
    "go": """
        (function_declaration name: (identifier) @function)
        (method_declaration name: (field_identifier) @method)
        (type_declaration (type_spec name: (type_identifier) @class))
    """,
    "rust": """
        (function_item name: (identifier) @function)
        (struct_item name: (type_identifier) @class)
        (enum_item name: (type_identifier) @class)
        (trait_item name: (type_identifier) @class)
    """,
    "java": """
        (class_declaration name: (identifier) @class)
        (interface_declaration name: (identifier) @class)
        (method_declaration name: (identifier) @method)
    """,
    "ruby": """
        (method name: (identifier) @method)
        (class name: (constant) @class)
        (module name: (constant) @class)
    """,
}

_TS_CACHE: Dict[str, Any] = {}


def _ts_tools(lang: str):
    """(parser, query) for a language, or None when tree-sitter (or the
    language pack) is unavailable. Cached; failures cache as None so a
    missing grammar is probed once."""
    if lang in _TS_CACHE:
        return _TS_CACHE[lang]
    tools = None
    try:
        import tree_sitter_languages as tsl
        parser = tsl.get_parser(lang)
        language = tsl.get_language(lang)
        query = language.query(_TS_QUERIES[lang])
        tools = (parser, query)
    except Exception as e:          # ImportError, unknown grammar, bad query
        logger.debug("tree-sitter unavailable for %s: %s", lang, e)
    _TS_CACHE[lang] = tools
    return tools


def _ts_captures(query, root) -> List[Tuple[Any, str]]:
    """Normalize the two tree-sitter query APIs: ≤0.21 returns
    [(node, name)], ≥0.22 returns {name: [nodes]}."""
    caps = query.captures(root)
    if isinstance(caps, dict):
        return [(node, name) for name, nodes in caps.items()
                for node in nodes]
    return list(caps)


def extract_symbols_treesitter(content: str, ext: str
                               ) -> Optional[List[Tuple[str, str]]]:
    """Symbols via tree-sitter for one file; None when unavailable."""
    lang = _TS_EXT_LANG.get(ext)
    if not lang:
        return None
    tools = _ts_tools(lang)
    if tools is None:
        return None
    parser, query = tools
    try:
        tree = parser.parse(content.encode("utf-8", errors="replace"))
        out: List[Tuple[str, str]] = []
        seen: Set[str] = set()
        for node, kind in _ts_captures(query, tree.root_node):
            text = node.text
            name = (text.decode("utf-8", errors="replace")
                    if isinstance(text, bytes) else str(text))
            if name and name not in seen:
                seen.add(name)
                out.append((name, kind))
        return out
    except Exception as e:
        logger.debug("tree-sitter parse failed (%s): %s", ext, e)
        return None


class RepoMapper:
    def __init__(self, root: str, max_files: int = 2000,
                 max_file_bytes: int = 512 * 1024,
                 exclude_patterns: Optional[List[str]] = None):
        self.root = os.path.abspath(root)
        self.max_files = max_files
        self.max_file_bytes = max_file_bytes
        # reference RepoMap/RepoSummary accept glob-style exclusions
        self.exclude_patterns = list(exclude_patterns or [])
        self._symbols: Optional[Dict[str, List[Tuple[str, str]]]] = None

    def _excluded(self, relpath: str) -> bool:
        import fnmatch
        return any(fnmatch.fnmatch(relpath, pat) or
                   fnmatch.fnmatch(os.path.basename(relpath), pat)
                   for pat in self.exclude_patterns)

    # -- walking & extraction ------------------------------------------------

    def source_files(self) -> List[str]:
        out: List[str] = []
        for dirpath, dirnames, filenames in os.walk(self.root):
            dirnames[:] = [d for d in dirnames if d not in EXCLUDE_DIRS and not d.startswith(".")]
            for fn in sorted(filenames):
                if os.path.splitext(fn)[1] in SOURCE_EXTS:
                    full = os.path.join(dirpath, fn)
                    if self.exclude_patterns and \
                            self._excluded(os.path.relpath(full, self.root)):
                        continue
                    try:
                        if os.path.getsize(full) <= self.max_file_bytes:
                            out.append(full)
                    except OSError:
                        pass
                if len(out) >= self.max_files:
                    return out
        return out

    @staticmethod
    def _read(path: str) -> str:
        try:
            with open(path, "r", encoding="utf-8", errors="replace") as f:
                return f.read()
        except OSError:
            return ""

    def extract_symbols(self, path: str) -> List[Tuple[str, str]]:
        """Return [(symbol, kind)] for one file."""
        content = self._read(path)
        if not content:
            return []
        if path.endswith(".py"):
            try:
                tree = ast.parse(content)
                out: List[Tuple[str, str]] = []
                for node in tree.body:
                    if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)):
                        out.append((node.name, "function"))
                    elif isinstance(node, ast.ClassDef):
                        out.append((node.name, "class"))
                        for sub in node.body:
                            if isinstance(sub, (ast.FunctionDef, ast.AsyncFunctionDef)):
                                out.append((f"{node.name}.{sub.name}", "method"))
                return out
            except SyntaxError:
                pass
        ts = extract_symbols_treesitter(content, os.path.splitext(path)[1])
        if ts:
            return ts
        out = []
        seen: Set[str] = set()
        for rx, kind in _REGEX_SYMBOLS:
            for m in rx.finditer(content):
                name = m.group(1)
                if name not in seen:
                    seen.add(name)
                    out.append((name, kind))
        return out

    def symbols(self) -> Dict[str, List[Tuple[str, str]]]:
        if self._symbols is None:
            files = self.source_files()
            with ThreadPoolExecutor(max_workers=8) as pool:
                results = pool.map(lambda p: (p, self.extract_symbols(p)), files)
            self._symbols = dict(results)
        return self._symbols

    # -- dependency graph & ranking ------------------------------------------

    def dependencies(self) -> Dict[str, List[str]]:
        """file -> imported module strings."""
        deps: Dict[str, List[str]] = {}
        for path in self.source_files():
            content = self._read(path)
            found: List[str] = []
            for rx in _IMPORT_RE:
                found.extend(rx.findall(content))
            if found:
                deps[os.path.relpath(path, self.root)] = sorted(set(found))
        return deps

    def _reference_graph(self) -> Dict[str, Set[str]]:
        """file -> set of files whose symbols it mentions
        (reference: repomap.py:391-421)."""
        syms = self.symbols()
        defined_in: Dict[str, str] = {}
        for path, symlist in syms.items():
            for name, _kind in symlist:
                base = name.split(".")[0]
                defined_in.setdefault(base, path)
        graph: Dict[str, Set[str]] = defaultdict(set)
        names = [n for n in defined_in if len(n) >= 4]
        if not names:
            return graph
        rx = re.compile(r"\b(" + "|".join(re.escape(n) for n in sorted(names, key=len, reverse=True)[:400]) + r")\b")
        for path in syms:
            content = self._read(path)
            for m in set(rx.findall(content)):
                target = defined_in.get(m)
                if target and target != path:
                    graph[path].add(target)
        return graph

    def _rank_files(self) -> List[Tuple[str, float]]:
        """In-degree weighted rank (PageRank-lite; reference: repomap.py:423-441)."""
        graph = self._reference_graph()
        score: Dict[str, float] = defaultdict(float)
        for path in self.symbols():
            score[path] += 0.1
        for src, targets in graph.items():
            for t in targets:
                score[t] += 1.0 / max(1, len(targets))
        return sorted(score.items(), key=lambda kv: kv[1], reverse=True)

    # -- output --------------------------------------------------------------

    def generate_map(self, token_budget: int = 2000) -> str:
        """Budgeted textual map: top-ranked files with their symbols.
        Token estimate: ~4 chars/token (reference: repomap.py:443-495)."""
        char_budget = token_budget * 4
        syms = self.symbols()
        parts: List[str] = []
        used = 0
        for path, _score in self._rank_files():
            rel = os.path.relpath(path, self.root)
            lines = [rel + ":"]
            for name, kind in syms.get(path, [])[:30]:
                lines.append(f"  {kind} {name}")
            block = "\n".join(lines) + "\n"
            if used + len(block) > char_budget:
                if used == 0:
                    parts.append(block[:char_budget])
                break
            parts.append(block)
            used += len(block)
        return "\n".join(parts)

    def generate_json(self) -> Dict[str, Any]:
        syms = self.symbols()
        return {
            "root": self.root,
            "files": {
                os.path.relpath(p, self.root): [{"name": n, "kind": k} for n, k in s]
                for p, s in syms.items()
            },
        }

    def summary(self) -> Dict[str, Any]:
        """File counts by language, largest files, top-level layout
        (reference RepoMapSummary: repomap.py:547-667)."""
        files = self.source_files()
        by_ext: Dict[str, int] = defaultdict(int)
        sizes: List[Tuple[str, int]] = []
        for p in files:
            by_ext[os.path.splitext(p)[1]] += 1
            try:
                sizes.append((os.path.relpath(p, self.root), os.path.getsize(p)))
            except OSError:
                pass
        sizes.sort(key=lambda kv: kv[1], reverse=True)
        top_level = sorted(
            d for d in os.listdir(self.root)
            if os.path.isdir(os.path.join(self.root, d)) and d not in EXCLUDE_DIRS
            and not d.startswith(".")
        ) if os.path.isdir(self.root) else []
        return {
            "root": self.root,
            "file_count": len(files),
            "by_extension": dict(by_ext),
            "largest_files": sizes[:10],
            "top_level_dirs": top_level,
        }


def generate_repo_map(path: str, token_budget: int = 2000) -> str:
    return RepoMapper(path).generate_map(token_budget=token_budget)
