"""memdir search: query language + evaluator.

Query-language parity with the reference (memdir_tools/search.py:392-519):

  shortcuts:      ``#tag`` (has_tag), ``+F`` (has_flag F), ``/regex/``
  field queries:  ``field:value`` (contains), ``field=value`` (equals),
                  ``field!=value``, ``field<value``, ``field>value``
  modifiers:      ``sort:field``, ``limit:N``, ``with_content``
  bare keywords:  OR-match across Subject + content (search.py:244-335)
  relative dates: ``now-7d`` / ``now+2w`` / ``now-1m`` / ``now-1y``
                  in date comparisons (search.py:201-225)

Fields resolve against headers first; ``status`` means the maildir status
dir unless a ``Status`` header exists (search.py:97-139, 315-318).

Performance note: the reference re-reads every file per query
(search.py:361-367). We keep the same full-scan lexical path for format
compatibility, but fei_amd.memdir.embed_index adds a persisted embedding
index (GPU GEMM top-k) for semantic search at corpus scale.
"""

from __future__ import annotations

import re
import time
from typing import Any, Dict, List, Optional, Tuple

from fei_amd.memdir import utils as mu

_REL_DATE_RE = re.compile(r"^now([+-])(\d+)([dwmy])$")
_UNIT_SECONDS = {"d": 86400, "w": 7 * 86400, "m": 30 * 86400, "y": 365 * 86400}


def _resolve_relative_date(expr: str) -> Optional[float]:
    m = _REL_DATE_RE.match(expr.strip().lower())
    if not m:
        return None
    sign = 1 if m.group(1) == "+" else -1
    return time.time() + sign * int(m.group(2)) * _UNIT_SECONDS[m.group(3)]


class SearchQuery:
    """Builder for a memdir query (reference: search.py:21-95)."""

    def __init__(self):
        self.conditions: List[Tuple[str, str, Any]] = []  # (field, op, value)
        self.keywords: List[str] = []
        self.folders: Optional[List[str]] = None
        self.statuses: Optional[List[str]] = None
        self.sort_field: Optional[str] = None
        self.sort_reverse: bool = True
        self.limit: Optional[int] = None
        self.offset: int = 0
        self.with_content: bool = False

    def add_condition(self, field: str, op: str, value: Any) -> "SearchQuery":
        self.conditions.append((field, op, value))
        return self

    def add_keyword(self, word: str) -> "SearchQuery":
        self.keywords.append(word)
        return self

    def set_folders(self, folders: List[str]) -> "SearchQuery":
        self.folders = folders
        return self

    def set_statuses(self, statuses: List[str]) -> "SearchQuery":
        self.statuses = statuses
        return self

    def set_sort(self, field: str, reverse: bool = True) -> "SearchQuery":
        self.sort_field = field
        self.sort_reverse = reverse
        return self

    def set_pagination(self, limit: Optional[int], offset: int = 0) -> "SearchQuery":
        self.limit = limit
        self.offset = offset
        return self


def _get_field_value(mem: Dict[str, Any], field: str) -> Any:
    """Resolve a field name against a memory (reference: search.py:97-139).
    Precedence: exact header -> case-insensitive header -> maildir attrs."""
    headers = mem.get("headers", {})
    # date/timestamp resolve to the maildir timestamp (numeric) even though a
    # Date header exists — comparisons and relative dates need the number.
    if field.lower() in ("date", "timestamp"):
        meta = mem.get("metadata") or {}
        return meta.get("timestamp")
    if field in headers:
        return headers[field]
    for k, v in headers.items():
        if k.lower() == field.lower():
            return v
    lf = field.lower()
    if lf == "status":
        return mem.get("status")
    if lf in ("date", "timestamp"):
        meta = mem.get("metadata") or {}
        return meta.get("timestamp")
    if lf == "flags":
        meta = mem.get("metadata") or {}
        return "".join(meta.get("flags", []))
    if lf in ("content", "body"):
        return mem.get("content", "")
    if lf == "folder":
        return mem.get("folder", "")
    if lf == "filename":
        return mem.get("filename", "")
    return None


def _compare_values(actual: Any, op: str, expected: Any) -> bool:
    """Operator evaluation (reference: search.py:141-242)."""
    if op == "has_flag":
        return isinstance(actual, str) and str(expected).upper() in actual.upper()
    if op == "has_tag":
        if not isinstance(actual, str):
            return False
        tags = [t.strip().lower() for t in re.split(r"[,\s]+", actual) if t.strip()]
        return str(expected).lower() in tags
    if actual is None:
        return op == "!=" and expected is not None
    if op == "matches":
        try:
            return re.search(str(expected), str(actual), re.IGNORECASE) is not None
        except re.error:
            return False
    if op == "contains":
        return str(expected).lower() in str(actual).lower()
    if op == "startswith":
        return str(actual).lower().startswith(str(expected).lower())
    if op == "endswith":
        return str(actual).lower().endswith(str(expected).lower())

    # =, !=, <, > with date/number awareness
    exp: Any = expected
    act: Any = actual
    rel = _resolve_relative_date(str(expected)) if isinstance(expected, str) else None
    if rel is not None:
        exp = rel
        try:
            act = float(actual)
        except (TypeError, ValueError):
            return False
    else:
        try:
            act_f, exp_f = float(actual), float(expected)
            act, exp = act_f, exp_f
        except (TypeError, ValueError):
            act, exp = str(actual).lower(), str(expected).lower()
    if op == "=":
        return act == exp
    if op == "!=":
        return act != exp
    if op == "<":
        return act < exp
    if op == ">":
        return act > exp
    return False


def _matches(mem: Dict[str, Any], query: SearchQuery) -> bool:
    for field, op, value in query.conditions:
        if not _compare_values(_get_field_value(mem, field), op, value):
            return False
    if query.keywords:
        # bare keywords OR across Subject + content (reference: search.py:244-335)
        hay = (str(_get_field_value(mem, "Subject") or "") + "\n" +
               str(mem.get("content", ""))).lower()
        if not any(kw.lower() in hay for kw in query.keywords):
            return False
    return True


def search_memories(
    query: SearchQuery,
    base: Optional[str] = None,
) -> List[Dict[str, Any]]:
    """Execute a query: full scan over folder x status
    (reference executor: search.py:337-390)."""
    folders = query.folders if query.folders is not None else mu.list_folders(base)
    statuses = query.statuses if query.statuses is not None else ["cur", "new"]
    results: List[Dict[str, Any]] = []
    for folder in folders:
        for status in statuses:
            for mem in mu.list_memories(folder, status, include_content=True, base=base):
                if _matches(mem, query):
                    results.append(mem)

    sort_field = query.sort_field or "date"
    def sort_key(m: Dict[str, Any]):
        v = _get_field_value(m, sort_field)
        if v is None:
            return (0, "")
        try:
            return (1, float(v))
        except (TypeError, ValueError):
            return (1, str(v).lower())
    try:
        results.sort(key=sort_key, reverse=query.sort_reverse)
    except TypeError:
        pass

    start = query.offset
    end = start + query.limit if query.limit else None
    results = results[start:end]
    if not query.with_content:
        for m in results:
            m.pop("content", None)
    return results


_FIELD_OP_RE = re.compile(r"^([A-Za-z_][A-Za-z0-9_]*)(!=|[:=<>])(.*)$")


def parse_search_args(query_string: str) -> SearchQuery:
    """Parse the query string language (reference: search.py:392-519)."""
    q = SearchQuery()
    # tokenize respecting quotes
    tokens: List[str] = []
    for m in re.finditer(r'"([^"]*)"|(/(?:[^/\\]|\\.)*/)|(\S+)', query_string):
        if m.group(1) is not None:
            tokens.append(m.group(1))
        elif m.group(2) is not None:
            tokens.append(m.group(2))
        else:
            tokens.append(m.group(3))
    for tok in tokens:
        if not tok:
            continue
        if tok.startswith("#") and len(tok) > 1:               # tag shortcut
            q.add_condition("Tags", "has_tag", tok[1:])
            continue
        if tok.startswith("+") and len(tok) == 2 and tok[1].upper() in mu.FLAGS:
            q.add_condition("flags", "has_flag", tok[1].upper())
            continue
        if tok.startswith("/") and tok.endswith("/") and len(tok) > 2:   # regex
            q.add_condition("content", "matches", tok[1:-1])
            continue
        if tok == "with_content":
            q.with_content = True
            continue
        m = _FIELD_OP_RE.match(tok)
        if m:
            field, op, value = m.group(1), m.group(2), m.group(3)
            if field.lower() == "sort":
                reverse = value.startswith("-")
                q.set_sort(value.lstrip("-+"), reverse=reverse or value in ("date", "timestamp"))
                continue
            if field.lower() == "limit":
                try:
                    q.limit = int(value)
                except ValueError:
                    pass
                continue
            if field.lower() == "offset":
                try:
                    q.offset = int(value)
                except ValueError:
                    pass
                continue
            if field.lower() == "folder":
                q.folders = (q.folders or []) + [value]
                continue
            if value.startswith("/") and value.endswith("/") and len(value) > 2:
                q.add_condition(field, "matches", value[1:-1])
            elif op == ":":
                q.add_condition(field, "contains", value)
            else:
                q.add_condition(field, op, value)
            continue
        q.add_keyword(tok)
    return q


def search(query_string: str, base: Optional[str] = None) -> List[Dict[str, Any]]:
    return search_memories(parse_search_args(query_string), base=base)


# -- output formatting (reference: search.py:521-594) ------------------------

def format_results(results: List[Dict[str, Any]], fmt: str = "text") -> str:
    if fmt == "json":
        import json
        return json.dumps(results, indent=2, default=str)
    if fmt == "csv":
        lines = ["filename,folder,status,subject"]
        for m in results:
            subj = (m.get("headers", {}).get("Subject", "") or "").replace(",", " ")
            lines.append(f"{m['filename']},{m.get('folder','')},{m.get('status','')},{subj}")
        return "\n".join(lines)
    if fmt == "compact":
        return "\n".join(
            f"{(m.get('metadata') or {}).get('unique','????????')} "
            f"[{m.get('folder') or 'Inbox'}] {m.get('headers', {}).get('Subject', '(no subject)')}"
            for m in results
        )
    lines = []
    for m in results:
        meta = m.get("metadata") or {}
        lines.append(f"ID: {meta.get('unique')}  Folder: {m.get('folder') or '(root)'}  "
                     f"Status: {m.get('status')}  Flags: {''.join(meta.get('flags', []))}")
        for k, v in m.get("headers", {}).items():
            lines.append(f"  {k}: {v}")
        if "content" in m:
            body = m["content"]
            lines.append("  " + (body[:200].replace("\n", "\n  ")))
        lines.append("")
    return "\n".join(lines)
