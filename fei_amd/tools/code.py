"""Filesystem / code tools behind the schemas in definitions.py.

Parity targets (behavior, not code): the reference's GlobFinder/GrepTool/
CodeEditor/FileViewer/DirectoryExplorer/SystemInfo/ShellRunner
(fei/tools/code.py:49-1714). Notable reference semantics preserved:
  - glob results cached briefly and path-jailed to a base path (code.py:60-83)
  - binary files detected by NUL-byte sniff and skipped (code.py:270-293)
  - grep scans files in a thread pool with a per-file match cap (code.py:408-500)
  - edits keep timestamped backups under ``.fei_backups`` capped at 10
    (code.py:524-616); Edit requires a unique match (code.py:618-668)
  - regex edit validates the result (Python ast) and rolls back (code.py:737-932)
  - Shell has an allowlist + denylist + interactive-command heuristic and a
    50 kB output cap (code.py:1352-1714)

Reference defect intentionally fixed (SURVEY.md known-defects list):
``Edit`` with an *empty* old_string creates the file (the reference's
handler checked ``None`` and could never create files).
"""

from __future__ import annotations

import ast
import fnmatch
import glob as globlib
import hashlib
import os
import re
import shlex
import shutil
import subprocess
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, List, Optional, Tuple

from fei_amd.utils.logging import get_logger

logger = get_logger("tools.code")

MAX_OUTPUT_BYTES = 50_000
MAX_MATCHES_PER_FILE = 1000
GLOB_CACHE_TTL_S = 60.0


def is_binary_file(path: str, sniff: int = 4096) -> bool:
    """Heuristic binary detection: NUL byte in the first 4 kB."""
    try:
        with open(path, "rb") as f:
            chunk = f.read(sniff)
        return b"\x00" in chunk
    except OSError:
        return True


class GlobFinder:
    """Glob matching with a short result cache and a base-path jail."""

    def __init__(self, base_path: Optional[str] = None):
        self.base_path = os.path.abspath(base_path) if base_path else None
        self._cache: Dict[Tuple[str, str], Tuple[float, List[str]]] = {}
        self._lock = threading.Lock()

    def _check_path_safety(self, path: str) -> str:
        resolved = os.path.abspath(path)
        if self.base_path and not (resolved + os.sep).startswith(self.base_path + os.sep) \
                and resolved != self.base_path:
            raise PermissionError(f"path {path!r} escapes the allowed base {self.base_path!r}")
        return resolved

    def find(self, pattern: str, path: Optional[str] = None) -> List[str]:
        root = self._check_path_safety(path or os.getcwd())
        key = (root, pattern)
        now = time.monotonic()
        with self._lock:
            hit = self._cache.get(key)
            if hit and now - hit[0] < GLOB_CACHE_TTL_S:
                return list(hit[1])
        matches = globlib.glob(os.path.join(root, pattern), recursive=True)
        files = [m for m in matches if os.path.isfile(m)]
        files.sort(key=lambda p: os.path.getmtime(p) if os.path.exists(p) else 0, reverse=True)
        with self._lock:
            self._cache[key] = (now, files)
        return files

    def batch(self, patterns: List[str], path: Optional[str] = None) -> Dict[str, List[str]]:
        with ThreadPoolExecutor(max_workers=min(8, max(1, len(patterns)))) as pool:
            results = pool.map(lambda p: (p, self.find(p, path)), patterns)
        return dict(results)


class GrepTool:
    """Parallel regex content search with a compiled-regex cache."""

    def __init__(self, finder: Optional[GlobFinder] = None):
        self.finder = finder or GlobFinder()
        self._regex_cache: Dict[Tuple[str, int], re.Pattern] = {}
        self._lock = threading.Lock()

    def _compile(self, pattern: str, flags: int = 0) -> re.Pattern:
        key = (pattern, flags)
        with self._lock:
            rx = self._regex_cache.get(key)
            if rx is None:
                rx = re.compile(pattern, flags)
                self._regex_cache[key] = rx
            return rx

    def search_single_file(self, path: str, rx: re.Pattern) -> List[Dict[str, Any]]:
        out: List[Dict[str, Any]] = []
        if is_binary_file(path):
            return out
        try:
            with open(path, "r", encoding="utf-8", errors="replace") as f:
                for lineno, line in enumerate(f, 1):
                    if rx.search(line):
                        out.append({"file": path, "line": lineno, "text": line.rstrip("\n")[:500]})
                        if len(out) >= MAX_MATCHES_PER_FILE:
                            break
        except OSError:
            pass
        return out

    def search(
        self,
        pattern: str,
        path: Optional[str] = None,
        include: Optional[str] = None,
    ) -> List[Dict[str, Any]]:
        rx = self._compile(pattern)
        root = os.path.abspath(path or os.getcwd())
        # 'include' is a basename glob applied anywhere under root (the
        # reference treated it as rooted — a known defect we do not copy).
        candidates: List[str] = []
        for dirpath, dirnames, filenames in os.walk(root):
            dirnames[:] = [d for d in dirnames if d not in (".git", "__pycache__", "node_modules")]
            for fn in filenames:
                if include and not fnmatch.fnmatch(fn, include):
                    continue
                candidates.append(os.path.join(dirpath, fn))
        results: List[Dict[str, Any]] = []
        with ThreadPoolExecutor(max_workers=8) as pool:
            for file_matches in pool.map(lambda p: self.search_single_file(p, rx), candidates):
                results.extend(file_matches)
        results.sort(key=lambda m: (m["file"], m["line"]))
        return results

    def find_in_files(self, files: List[str], pattern: str,
                      case_sensitive: bool = False) -> Dict[str, List[Dict[str, Any]]]:
        # reference default: case-insensitive (definitions.py FindInFiles)
        rx = self._compile(pattern, 0 if case_sensitive else re.IGNORECASE)
        out: Dict[str, List[Dict[str, Any]]] = {}
        for path in files:
            if not os.path.isfile(path):
                out[path] = [{"error": "not a file"}]
                continue
            out[path] = self.search_single_file(path, rx)
        return out


class CodeEditor:
    """File editing with backups, unique-match edits and validated regex edits."""

    BACKUP_DIR = ".fei_backups"
    MAX_BACKUPS = 10

    def _backup(self, file_path: str) -> Optional[str]:
        if not os.path.exists(file_path):
            return None
        d = os.path.join(os.path.dirname(os.path.abspath(file_path)), self.BACKUP_DIR)
        os.makedirs(d, exist_ok=True)
        stamp = time.strftime("%Y%m%d-%H%M%S")
        dest = os.path.join(d, f"{os.path.basename(file_path)}.{stamp}.{os.getpid()}")
        shutil.copy2(file_path, dest)
        # cap backups per file
        base = os.path.basename(file_path) + "."
        backups = sorted(p for p in os.listdir(d) if p.startswith(base))
        for old in backups[: max(0, len(backups) - self.MAX_BACKUPS)]:
            try:
                os.unlink(os.path.join(d, old))
            except OSError:
                pass
        return dest

    def edit_file(self, file_path: str, old_string: str, new_string: str) -> Dict[str, Any]:
        if old_string == "":
            return self.create_file(file_path, new_string)
        if not os.path.isfile(file_path):
            return {"error": f"file not found: {file_path}"}
        with open(file_path, "r", encoding="utf-8") as f:
            content = f.read()
        count = content.count(old_string)
        if count == 0:
            return {"error": "old_string not found in file"}
        if count > 1:
            return {"error": f"old_string occurs {count} times; it must be unique — add context"}
        self._backup(file_path)
        new_content = content.replace(old_string, new_string, 1)
        with open(file_path, "w", encoding="utf-8") as f:
            f.write(new_content)
        return {"success": True, "file": file_path, "message": "edit applied"}

    def create_file(self, file_path: str, content: str) -> Dict[str, Any]:
        if os.path.exists(file_path):
            return {"error": f"file already exists: {file_path} (use Replace to overwrite)"}
        parent = os.path.dirname(os.path.abspath(file_path))
        os.makedirs(parent, exist_ok=True)
        with open(file_path, "w", encoding="utf-8") as f:
            f.write(content)
        return {"success": True, "file": file_path, "message": "file created"}

    def replace_file(self, file_path: str, content: str) -> Dict[str, Any]:
        if os.path.exists(file_path):
            self._backup(file_path)
        parent = os.path.dirname(os.path.abspath(file_path))
        if parent:
            os.makedirs(parent, exist_ok=True)
        with open(file_path, "w", encoding="utf-8") as f:
            f.write(content)
        return {"success": True, "file": file_path, "message": "file written"}

    # -- post-edit validators (reference code.py:827-932: ast, esprima,
    # pylint, flake8 — each optional, skipped gracefully when absent) ------

    @staticmethod
    def _v_ast(content: str) -> Optional[str]:
        try:
            ast.parse(content)
        except SyntaxError as e:
            return f"python syntax error after edit: {e}"
        return None

    @staticmethod
    def _v_pyflakes(content: str) -> Optional[str]:
        try:
            from io import StringIO

            from pyflakes.api import check
            from pyflakes.reporter import Reporter
        except ImportError:
            return None                     # optional linter absent: skip
        out, err = StringIO(), StringIO()
        n = check(content, "<edit>", Reporter(out, err))
        hard = [ln for ln in (out.getvalue() + err.getvalue()).splitlines()
                if "undefined name" in ln or "syntax" in ln.lower()]
        if n and hard:
            return "pyflakes: " + "; ".join(hard[:3])
        return None

    @staticmethod
    def _v_esprima(content: str) -> Optional[str]:
        try:
            import esprima
        except ImportError:
            return None
        try:
            esprima.parseScript(content, tolerant=False)
        except Exception as e:
            return f"javascript syntax error after edit: {e}"
        return None

    @staticmethod
    def _v_json(content: str) -> Optional[str]:
        import json as _json
        try:
            _json.loads(content)
        except ValueError as e:
            return f"json syntax error after edit: {e}"
        return None

    @staticmethod
    def _v_yaml(content: str) -> Optional[str]:
        try:
            import yaml
        except ImportError:
            return None
        try:
            yaml.safe_load(content)
        except Exception as e:
            return f"yaml syntax error after edit: {e}"
        return None

    @staticmethod
    def _v_treesitter(content: str, ext: str) -> Optional[str]:
        """Generic syntax gate for any language with a grammar: parse and
        reject when the tree contains ERROR nodes. Skipped when
        tree-sitter is unavailable (optional in this image)."""
        from fei_amd.tools.repomap import _TS_EXT_LANG, _ts_tools
        lang = _TS_EXT_LANG.get(ext)
        if not lang:
            return None
        tools = _ts_tools(lang)
        if tools is None:
            return None
        try:
            tree = tools[0].parse(content.encode("utf-8", errors="replace"))
            root = tree.root_node
            if getattr(root, "has_error", False):
                return f"{lang} syntax error after edit (tree-sitter)"
        except Exception:
            return None
        return None

    VALIDATORS = {
        "ast": "_v_ast", "pyflakes": "_v_pyflakes", "esprima": "_v_esprima",
        "json": "_v_json", "yaml": "_v_yaml", "treesitter": "_v_treesitter",
    }
    _EXT_VALIDATORS = {
        ".py": ["ast", "pyflakes"],
        ".js": ["esprima", "treesitter"], ".jsx": ["esprima", "treesitter"],
        ".ts": ["treesitter"], ".tsx": ["treesitter"],
        ".json": ["json"],
        ".yaml": ["yaml"], ".yml": ["yaml"],
        ".c": ["treesitter"], ".h": ["treesitter"], ".cpp": ["treesitter"],
        ".hpp": ["treesitter"], ".cc": ["treesitter"], ".go": ["treesitter"],
        ".rs": ["treesitter"], ".java": ["treesitter"], ".rb": ["treesitter"],
    }

    def _validate_code(self, file_path: str, content: str,
                       validators: Optional[List[str]] = None
                       ) -> Optional[str]:
        """Post-edit validation; returns an error message or None. The
        validator set defaults by extension; optional checkers (pyflakes,
        esprima, yaml, tree-sitter) skip silently when not importable —
        the reference's esprima/pylint/flake8 behavior (code.py:827-932)."""
        ext = os.path.splitext(file_path)[1]
        names = validators if validators is not None else \
            self._EXT_VALIDATORS.get(ext, [])
        for name in names:
            attr = self.VALIDATORS.get(name)
            if not attr:
                continue
            fn = getattr(self, attr)
            err = fn(content, ext) if name == "treesitter" else fn(content)
            if err:
                return err
        return None

    def regex_replace(
        self,
        file_path: str,
        pattern: str,
        replacement: str,
        count: int = 0,
        validate: bool = True,
        validators: Optional[List[str]] = None,
    ) -> Dict[str, Any]:
        """``validators``: explicit validator list (reference semantics:
        ['ast'] forces the python check, [] disables). None = by file
        extension, gated on ``validate``."""
        if not os.path.isfile(file_path):
            return {"error": f"file not found: {file_path}"}
        with open(file_path, "r", encoding="utf-8") as f:
            content = f.read()
        try:
            rx = re.compile(pattern, re.MULTILINE)
        except re.error as e:
            return {"error": f"bad regex: {e}"}
        new_content, n = rx.subn(replacement, content, count=count)
        if n == 0:
            return {"error": "pattern did not match"}
        if validate or validators is not None:
            # validators=None: defaults by extension; []: disabled;
            # explicit list: exactly those (reference semantics)
            err = self._validate_code(file_path, new_content,
                                      validators=validators)
            if err:
                return {"error": err, "replacements": 0}
        backup = self._backup(file_path)
        with open(file_path, "w", encoding="utf-8") as f:
            f.write(new_content)
        return {"success": True, "file": file_path, "replacements": n, "backup": backup}


class FileViewer:
    """Read files with offset/limit, line counts and hashes."""

    def view(self, file_path: str, offset: int = 1, limit: int = 2000) -> Dict[str, Any]:
        if not os.path.isfile(file_path):
            return {"error": f"file not found: {file_path}"}
        if is_binary_file(file_path):
            size = os.path.getsize(file_path)
            return {"error": f"binary file ({size} bytes): {file_path}"}
        offset = max(1, offset)
        lines: List[str] = []
        total = 0
        with open(file_path, "r", encoding="utf-8", errors="replace") as f:
            for i, line in enumerate(f, 1):
                total = i
                if i >= offset and len(lines) < limit:
                    lines.append(f"{i}\t{line.rstrip(chr(10))}")
        return {
            "file": file_path,
            "offset": offset,
            "lines_shown": len(lines),
            "total_lines": total,
            "content": "\n".join(lines),
            "truncated": total > offset - 1 + len(lines),
        }

    def count_lines(self, file_path: str) -> int:
        n = 0
        with open(file_path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                n += chunk.count(b"\n")
        return n

    def file_hash(self, file_path: str, algo: str = "sha256") -> str:
        h = hashlib.new(algo)
        with open(file_path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        return h.hexdigest()


class DirectoryExplorer:
    """Directory listings with ignore patterns."""

    def list_directory(
        self,
        path: str,
        ignore: Optional[List[str]] = None,
        recursive: bool = False,
        max_entries: int = 2000,
    ) -> Dict[str, Any]:
        if not os.path.isdir(path):
            return {"error": f"not a directory: {path}"}
        ignore = ignore or []

        def ignored(name: str) -> bool:
            return any(fnmatch.fnmatch(name, pat) for pat in ignore)

        entries: List[Dict[str, Any]] = []
        if recursive:
            for dirpath, dirnames, filenames in os.walk(path):
                dirnames[:] = [d for d in dirnames if not ignored(d)]
                for name in sorted(filenames):
                    if ignored(name):
                        continue
                    full = os.path.join(dirpath, name)
                    try:
                        entries.append({"path": os.path.relpath(full, path), "type": "file",
                                        "size": os.path.getsize(full)})
                    except OSError:
                        pass
                    if len(entries) >= max_entries:
                        return {"path": path, "entries": entries, "truncated": True}
        else:
            for name in sorted(os.listdir(path)):
                if ignored(name):
                    continue
                full = os.path.join(path, name)
                try:
                    if os.path.isdir(full):
                        entries.append({"path": name + "/", "type": "dir",
                                        "entries": len(os.listdir(full))})
                    else:
                        entries.append({"path": name, "type": "file", "size": os.path.getsize(full)})
                except OSError:
                    pass
                if len(entries) >= max_entries:
                    return {"path": path, "entries": entries, "truncated": True}
        return {"path": path, "entries": entries, "truncated": False}


class SystemInfo:
    """Host information (os/memory/disk), psutil-backed where available."""

    def info(self) -> Dict[str, Any]:
        import platform

        out: Dict[str, Any] = {
            "platform": platform.platform(),
            "python": platform.python_version(),
            "cpu_count": os.cpu_count(),
        }
        try:
            import psutil

            vm = psutil.virtual_memory()
            du = psutil.disk_usage("/")
            out["memory"] = {"total": vm.total, "available": vm.available}
            out["disk"] = {"total": du.total, "free": du.free}
        except Exception:  # noqa: BLE001 - psutil optional
            pass
        try:
            import torch

            out["gpu"] = {
                "available": torch.cuda.is_available(),
                "count": torch.cuda.device_count() if torch.cuda.is_available() else 0,
                "name": torch.cuda.get_device_name(0) if torch.cuda.is_available() else None,
            }
        except Exception:  # noqa: BLE001
            pass
        return out


class ShellRunner:
    """Guarded shell execution: allowlist + denylist + output caps.

    Reference parity: fei/tools/code.py:1348-1714 (allowlist 1352-1385,
    denylist 1388-1404, interactive heuristic 1494-1519, background mode
    1573-1665, foreground run with timeout + truncation 1667-1714).
    """

    ALLOWED_COMMANDS = {
        "ls", "cat", "head", "tail", "wc", "grep", "find", "echo", "pwd", "cd",
        "python", "python3", "pip", "pip3", "pytest", "git", "make", "cmake",
        "ninja", "gcc", "g++", "hipcc", "rocm-smi", "rocminfo", "rocprofv3",
        "sed", "awk", "sort", "uniq", "diff", "cp", "mv", "mkdir", "touch",
        "which", "env", "date", "du", "df", "tar", "gzip", "gunzip", "xargs",
        "true", "false", "sleep", "timeout", "nproc", "uname", "basename",
        "dirname", "realpath", "readlink", "stat", "file", "tr", "cut", "tee",
        "curl", "wget", "sh", "bash",
    }
    DENY_PATTERNS = [
        r"\brm\s+-rf\s+/(?:\s|$)",
        r"\bmkfs\b",
        r"\bdd\s+if=.*of=/dev/",
        r":\(\)\s*\{\s*:\|:",            # fork bomb
        r"\bshutdown\b|\breboot\b",
        r">\s*/dev/sd",
        r"\bchmod\s+-R\s+777\s+/(?:\s|$)",
    ]
    INTERACTIVE_COMMANDS = {"vi", "vim", "nano", "emacs", "less", "more", "top", "htop", "ssh"}

    def __init__(self):
        self._lock = threading.RLock()
        self._background: Dict[int, subprocess.Popen] = {}

    def _check_command(self, command: str) -> Optional[str]:
        for pat in self.DENY_PATTERNS:
            if re.search(pat, command):
                return f"command denied (matches destructive pattern {pat!r})"
        try:
            tokens = shlex.split(command)
        except ValueError as e:
            return f"cannot parse command: {e}"
        if not tokens:
            return "empty command"
        # Check the first token of each pipeline stage / sequence element.
        heads: List[str] = []
        expect_head = True
        for tok in tokens:
            if tok in ("|", "&&", "||", ";"):
                expect_head = True
                continue
            if expect_head:
                if "=" in tok and not tok.startswith("="):  # env assignment prefix
                    continue
                heads.append(os.path.basename(tok))
                expect_head = False
        for head in heads:
            if head in self.INTERACTIVE_COMMANDS:
                return f"interactive command not supported: {head}"
            if head not in self.ALLOWED_COMMANDS:
                return f"command not in allowlist: {head}"
        return None

    def run(
        self,
        command: str,
        timeout: float = 60.0,
        background: bool = False,
        working_dir: Optional[str] = None,
    ) -> Dict[str, Any]:
        err = self._check_command(command)
        if err:
            return {"error": err}
        if working_dir and not os.path.isdir(working_dir):
            return {"error": f"working_dir does not exist: {working_dir}"}
        if background:
            proc = subprocess.Popen(
                command, shell=True, cwd=working_dir,
                stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
                start_new_session=True,
            )
            with self._lock:
                self._background[proc.pid] = proc
            # timer kill so background jobs cannot run forever
            t = threading.Timer(timeout, self._kill_background, args=(proc.pid,))
            t.daemon = True
            t.start()
            return {"success": True, "pid": proc.pid, "background": True}
        try:
            result = subprocess.run(
                command, shell=True, cwd=working_dir, capture_output=True,
                timeout=timeout, text=True,
            )
        except subprocess.TimeoutExpired:
            return {"error": f"command timed out after {timeout}s"}
        stdout = result.stdout or ""
        stderr = result.stderr or ""
        if len(stdout) > MAX_OUTPUT_BYTES:
            stdout = stdout[:MAX_OUTPUT_BYTES] + "\n...[truncated]"
        if len(stderr) > MAX_OUTPUT_BYTES:
            stderr = stderr[:MAX_OUTPUT_BYTES] + "\n...[truncated]"
        return {
            "success": result.returncode == 0,
            "returncode": result.returncode,
            "stdout": stdout,
            "stderr": stderr,
        }

    def _kill_background(self, pid: int) -> None:
        with self._lock:
            proc = self._background.pop(pid, None)
        if proc and proc.poll() is None:
            try:
                proc.kill()
            except OSError:
                pass


# Module-level singletons (reference: code.py:1717-1724).
glob_finder = GlobFinder()
grep_tool = GrepTool(glob_finder)
code_editor = CodeEditor()
file_viewer = FileViewer()
directory_explorer = DirectoryExplorer()
system_info = SystemInfo()
shell_runner = ShellRunner()


def create_code_tools(registry) -> None:
    """Register the 14 code tools on a ToolRegistry
    (reference: code.py:1727-1866)."""
    from fei_amd.tools import definitions as d
    from fei_amd.tools import handlers as h

    pairs = [
        (d.GLOB_TOOL, h.glob_tool_handler),
        (d.GREP_TOOL, h.grep_tool_handler),
        (d.VIEW_TOOL, h.view_handler),
        (d.EDIT_TOOL, h.edit_handler),
        (d.REPLACE_TOOL, h.replace_handler),
        (d.LS_TOOL, h.ls_handler),
        (d.REGEX_EDIT_TOOL, h.regex_edit_handler),
        (d.BATCH_GLOB_TOOL, h.batch_glob_handler),
        (d.FIND_IN_FILES_TOOL, h.find_in_files_handler),
        (d.SMART_SEARCH_TOOL, h.smart_search_handler),
        (d.REPO_MAP_TOOL, h.repo_map_handler),
        (d.REPO_SUMMARY_TOOL, h.repo_summary_handler),
        (d.REPO_DEPS_TOOL, h.repo_deps_handler),
        (d.SHELL_TOOL, h.shell_handler),
    ]
    for spec, handler in pairs:
        registry.register_tool(spec["name"], spec["description"], spec["input_schema"], handler)
