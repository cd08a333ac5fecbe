"""Maildir-style primitives: the on-disk memory format.

Format contract (byte-compatible with the reference, memdir_tools/utils.py):
  - base dir default ``./Memdir`` (utils.py:16), overridable by env
    ``MEMDIR_BASE`` or config ``memdir.base``
  - per-folder status dirs ``cur`` / ``new`` / ``tmp`` (utils.py:19)
  - special folders ``.Trash`` ``.ToDoLater`` ``.Projects`` ``.Archive``
    (utils.py:22)
  - flags: S=Seen R=Replied F=Flagged P=Priority (utils.py:25-30)
  - filename grammar ``{unix_ts}.{uuid4hex[:8]}.{hostname}:2,{FLAGS}``
    (utils.py:59-72), parsed by regex (utils.py:74-95)
  - content: ``Header: value`` lines, a ``---`` separator line, then the
    body (utils.py:97-132)
  - atomic create: write into ``tmp`` then rename into ``new``
    (utils.py:153-200)
"""

from __future__ import annotations

import os
import re
import socket
import time
import uuid
from typing import Any, Dict, List, Optional, Tuple

STATUS_DIRS = ["cur", "new", "tmp"]
SPECIAL_FOLDERS = [".Trash", ".ToDoLater", ".Projects", ".Archive"]
FLAGS = {
    "S": "Seen",
    "R": "Replied",
    "F": "Flagged",
    "P": "Priority",
}

_FILENAME_RE = re.compile(
    r"^(?P<timestamp>\d+)\.(?P<unique>[0-9a-f]{8})\.(?P<hostname>[^:]+):2,(?P<flags>[A-Z]*)$"
)

HEADER_SEPARATOR = "---"


def get_memdir_base(base: Optional[str] = None) -> str:
    if base:
        return os.path.abspath(base)
    env = os.environ.get("MEMDIR_BASE")
    if env:
        return os.path.abspath(env)
    return os.path.abspath("Memdir")


def folder_path(folder: str = "", base: Optional[str] = None) -> str:
    """Resolve a folder name to its path under the memdir base, REJECTING
    names whose normalized path escapes the base. Folder names arrive from
    the HTTP server (unauthenticated by default), so '../../x' must never
    reach a filesystem operation. normpath (not realpath)
    so the folder manager's intentional in-tree symlinks keep working."""
    root = get_memdir_base(base)
    if not folder:
        return root
    p = os.path.normpath(os.path.join(root, folder))
    if p != root and not p.startswith(root + os.sep):
        raise ValueError(f"folder name escapes the memdir base: {folder!r}")
    return p


def check_status(status: str) -> str:
    """Statuses are exactly the maildir trio; anything else (e.g. a
    traversal attempt through the status field) is rejected."""
    if status not in STATUS_DIRS:
        raise ValueError(f"invalid status {status!r} (expected cur/new/tmp)")
    return status


def check_filename(filename: str) -> str:
    """Reject filenames containing path separators or dot-dot."""
    if (not filename or filename in (".", "..") or "/" in filename
            or "\\" in filename or os.sep in filename):
        raise ValueError(f"invalid memory filename: {filename!r}")
    return filename


def ensure_folder(folder: str = "", base: Optional[str] = None) -> str:
    """Create the folder (and its cur/new/tmp) if needed; return its path.
    ``folder`` is '' for the root folder, or a dotted name like '.Projects'
    or a path like '.Projects/python'."""
    fpath = folder_path(folder, base)
    for status in STATUS_DIRS:
        os.makedirs(os.path.join(fpath, status), exist_ok=True)
    return fpath


def generate_filename(flags: str = "", timestamp: Optional[float] = None) -> str:
    ts = int(timestamp if timestamp is not None else time.time())
    unique = uuid.uuid4().hex[:8]
    hostname = socket.gethostname().split(".")[0] or "localhost"
    hostname = hostname.replace(":", "_").replace(",", "_").replace("/", "_")
    flags = "".join(sorted(set(c for c in flags.upper() if c in FLAGS)))
    return f"{ts}.{unique}.{hostname}:2,{flags}"


def parse_memory_filename(filename: str) -> Optional[Dict[str, Any]]:
    m = _FILENAME_RE.match(filename)
    if not m:
        return None
    return {
        "timestamp": int(m.group("timestamp")),
        "unique": m.group("unique"),
        "hostname": m.group("hostname"),
        "flags": list(m.group("flags")),
    }


def format_memory_content(headers: Dict[str, str], body: str) -> str:
    lines = [f"{k}: {v}" for k, v in headers.items()]
    return "\n".join(lines) + f"\n{HEADER_SEPARATOR}\n" + body


def parse_memory_content(content: str) -> Tuple[Dict[str, str], str]:
    """Split ``Header: value`` lines + '---' + body. A file without the
    separator is all body (reference tolerates this, utils.py:97-132)."""
    headers: Dict[str, str] = {}
    lines = content.split("\n")
    body_start = None
    for i, line in enumerate(lines):
        if line.strip() == HEADER_SEPARATOR:
            body_start = i + 1
            break
        if ":" in line:
            key, _, value = line.partition(":")
            key = key.strip()
            if key and re.match(r"^[A-Za-z][A-Za-z0-9_-]*$", key):
                headers[key] = value.strip()
                continue
        # a non-header line before the separator: treat everything as body
        if line.strip():
            return {}, content
    if body_start is None:
        return headers, ""
    return headers, "\n".join(lines[body_start:])


def create_memory(
    folder: str = "",
    headers: Optional[Dict[str, str]] = None,
    body: str = "",
    flags: str = "",
    base: Optional[str] = None,
    status: str = "new",
) -> str:
    """Atomically create a memory file: write to tmp, rename into ``status``
    (default ``new``; reference: utils.py:153-200). Returns the filename."""
    headers = dict(headers or {})
    headers.setdefault("Date", time.strftime("%a, %d %b %Y %H:%M:%S +0000", time.gmtime()))
    fpath = ensure_folder(folder, base)
    check_status(status)
    filename = generate_filename(flags)
    tmp_path = os.path.join(fpath, "tmp", filename)
    final_path = os.path.join(fpath, status, filename)
    with open(tmp_path, "w", encoding="utf-8") as f:
        f.write(format_memory_content(headers, body))
        f.flush()
        os.fsync(f.fileno())
    os.rename(tmp_path, final_path)
    return filename


def _iter_status_files(folder_path: str, status: str) -> List[str]:
    d = os.path.join(folder_path, status)
    if not os.path.isdir(d):
        return []
    return sorted(os.listdir(d))


def list_memories(
    folder: str = "",
    status: str = "cur",
    include_content: bool = False,
    base: Optional[str] = None,
) -> List[Dict[str, Any]]:
    """List memories in one folder+status (reference: utils.py:202-253)."""
    fpath = folder_path(folder, base)
    check_status(status)
    out: List[Dict[str, Any]] = []
    for filename in _iter_status_files(fpath, status):
        meta = parse_memory_filename(filename)
        if meta is None:
            continue
        item: Dict[str, Any] = {
            "filename": filename,
            "folder": folder,
            "status": status,
            "metadata": meta,
        }
        path = os.path.join(fpath, status, filename)
        if include_content:
            try:
                with open(path, "r", encoding="utf-8", errors="replace") as f:
                    content = f.read()
                headers, body = parse_memory_content(content)
                item["headers"] = headers
                item["content"] = body
            except OSError:
                continue
        out.append(item)
    return out


def list_folders(base: Optional[str] = None) -> List[str]:
    """All folders (recursively), '' for root."""
    root = get_memdir_base(base)
    if not os.path.isdir(root):
        return []
    out = [""]
    for dirpath, dirnames, _ in os.walk(root):
        dirnames[:] = [d for d in dirnames if d not in STATUS_DIRS]
        for d in dirnames:
            out.append(os.path.relpath(os.path.join(dirpath, d), root))
    return sorted(set(out))


def find_memory(
    memory_id: str, base: Optional[str] = None,
    folder: Optional[str] = None,
) -> Optional[Tuple[str, str, str]]:
    """Locate a memory by full filename or unique-id prefix, optionally
    within one folder. Returns (folder, status, filename) or None."""
    folders = [folder] if folder is not None else list_folders(base)
    for folder in folders:
        fpath = folder_path(folder, base)
        for status in STATUS_DIRS:
            for filename in _iter_status_files(fpath, status):
                meta = parse_memory_filename(filename)
                if meta is None:
                    continue
                if filename == memory_id or meta["unique"] == memory_id or \
                        meta["unique"].startswith(memory_id):
                    return folder, status, filename
    return None


def read_memory(
    folder: str, status: str, filename: str, base: Optional[str] = None
) -> Optional[Dict[str, Any]]:
    path = os.path.join(folder_path(folder, base), check_status(status),
                        check_filename(filename))
    try:
        with open(path, "r", encoding="utf-8", errors="replace") as f:
            content = f.read()
    except OSError:
        return None
    headers, body = parse_memory_content(content)
    return {
        "filename": filename,
        "folder": folder,
        "status": status,
        "metadata": parse_memory_filename(filename),
        "headers": headers,
        "content": body,
    }


def move_memory(
    filename: str,
    src_folder: str,
    dst_folder: str,
    src_status: str = "new",
    dst_status: str = "cur",
    new_flags: Optional[str] = None,
    base: Optional[str] = None,
) -> bool:
    """Move by rename, optionally rewriting the flags part of the filename
    (reference: utils.py:255-297)."""
    src = os.path.join(folder_path(src_folder, base),
                       check_status(src_status), check_filename(filename))
    if not os.path.exists(src):
        return False
    new_name = filename
    if new_flags is not None:
        base_part = filename.split(":2,")[0]
        flags = "".join(sorted(set(c for c in new_flags.upper() if c in FLAGS)))
        new_name = f"{base_part}:2,{flags}"
    ensure_folder(dst_folder, base)
    dst = os.path.join(folder_path(dst_folder, base),
                       check_status(dst_status), new_name)
    os.rename(src, dst)
    return True


def update_memory_flags(
    filename: str,
    folder: str,
    status: str,
    flags: str,
    base: Optional[str] = None,
) -> Optional[str]:
    """Rewrite flags via rename in place (reference: utils.py:354-388).
    Returns the new filename or None."""
    fpath = folder_path(folder, base)
    src = os.path.join(fpath, check_status(status), check_filename(filename))
    if not os.path.exists(src):
        return None
    base_part = filename.split(":2,")[0]
    norm = "".join(sorted(set(c for c in flags.upper() if c in FLAGS)))
    new_name = f"{base_part}:2,{norm}"
    os.rename(src, os.path.join(fpath, status, new_name))
    return new_name


def search_memories_simple(
    query: str,
    folders: Optional[List[str]] = None,
    statuses: Optional[List[str]] = None,
    base: Optional[str] = None,
) -> List[Dict[str, Any]]:
    """Naive substring search over headers+body (reference: utils.py:299-352).
    The full query language lives in fei_amd.memdir.search."""
    q = query.lower()
    out: List[Dict[str, Any]] = []
    for folder in folders if folders is not None else list_folders(base):
        for status in statuses if statuses is not None else ["cur", "new"]:
            for mem in list_memories(folder, status, include_content=True, base=base):
                hay = "\n".join(f"{k}: {v}" for k, v in mem.get("headers", {}).items())
                hay = (hay + "\n" + mem.get("content", "")).lower()
                if q in hay:
                    out.append(mem)
    return out
