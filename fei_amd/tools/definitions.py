"""Tool schemas exposed to the model.

Parity: the reference ships 14 JSON-schema tool specs plus a search variant
(fei/tools/definitions.py:11-441). Same tool names and argument shapes so
transcripts/agents stay compatible; wording is our own.
"""

from __future__ import annotations

from typing import Any, Dict, List

GLOB_TOOL: Dict[str, Any] = {
    "name": "GlobTool",
    "description": (
        "Find files by glob pattern (e.g. '**/*.py', 'src/**/*.ts'). Returns "
        "matching paths sorted by modification time (newest first). Use this "
        "to locate files by name or extension."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "pattern": {"type": "string", "description": "Glob pattern to match against file paths"},
            "path": {"type": "string", "description": "Directory to search in (defaults to cwd)"},
        },
        "required": ["pattern"],
    },
}

GREP_TOOL: Dict[str, Any] = {
    "name": "GrepTool",
    "description": (
        "Search file contents with a regular expression. Returns matching "
        "lines with file path and line number. Use 'include' to filter which "
        "files are scanned (glob, e.g. '*.py')."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "pattern": {"type": "string", "description": "Regular expression to search for"},
            "path": {"type": "string", "description": "Directory to search in (defaults to cwd)"},
            "include": {"type": "string", "description": "Glob filter for files to scan (e.g. '*.py')"},
        },
        "required": ["pattern"],
    },
}

VIEW_TOOL: Dict[str, Any] = {
    "name": "View",
    "description": (
        "Read a file from the filesystem. Supports an optional byte-safe "
        "line offset and limit for large files. Lines are numbered in the "
        "output starting at the offset."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "file_path": {"type": "string", "description": "Absolute or relative path of the file to read"},
            "offset": {"type": "integer", "description": "1-based line number to start from"},
            "limit": {"type": "integer", "description": "Maximum number of lines to return"},
        },
        "required": ["file_path"],
    },
}

EDIT_TOOL: Dict[str, Any] = {
    "name": "Edit",
    "description": (
        "Edit a file by exact string replacement. 'old_string' must occur "
        "exactly once in the file; include enough context to make it unique. "
        "Pass an empty 'old_string' to create a new file with 'new_string' "
        "as its content."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "file_path": {"type": "string", "description": "Path of the file to edit"},
            "old_string": {"type": "string", "description": "Exact text to replace (empty to create a file)"},
            "new_string": {"type": "string", "description": "Replacement text"},
        },
        "required": ["file_path", "old_string", "new_string"],
    },
}

REPLACE_TOOL: Dict[str, Any] = {
    "name": "Replace",
    "description": "Overwrite a file with entirely new content (creates it if missing).",
    "input_schema": {
        "type": "object",
        "properties": {
            "file_path": {"type": "string", "description": "Path of the file to write"},
            "content": {"type": "string", "description": "New file content"},
        },
        "required": ["file_path", "content"],
    },
}

LS_TOOL: Dict[str, Any] = {
    "name": "LS",
    "description": "List files and directories at a path, with sizes; optionally ignore glob patterns.",
    "input_schema": {
        "type": "object",
        "properties": {
            "path": {"type": "string", "description": "Directory to list"},
            "ignore": {"type": "array", "items": {"type": "string"}, "description": "Glob patterns to skip"},
        },
        "required": ["path"],
    },
}

REGEX_EDIT_TOOL: Dict[str, Any] = {
    "name": "RegexEdit",
    "description": (
        "Edit a file by regex substitution (Python re syntax, applied with "
        "re.sub over the whole file). The edit is validated afterwards for "
        "Python files (ast.parse); on validation failure it is rolled back."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "file_path": {"type": "string", "description": "Path of the file to edit"},
            "pattern": {"type": "string", "description": "Regular expression to match"},
            "replacement": {"type": "string", "description": "Replacement text (may use backrefs like \\1)"},
            "count": {"type": "integer", "description": "Max replacements (0 = all)"},
            "validate": {"type": "boolean", "description": "Validate the result (default true)"},
            "validators": {"type": "array", "items": {"type": "string"}, "description": "Explicit validators (e.g. ['ast']); overrides auto-detection"},
        },
        "required": ["file_path", "pattern", "replacement"],
    },
}

BATCH_GLOB_TOOL: Dict[str, Any] = {
    "name": "BatchGlob",
    "description": "Run several glob patterns in one call; returns a mapping pattern -> matching files.",
    "input_schema": {
        "type": "object",
        "properties": {
            "patterns": {"type": "array", "items": {"type": "string"}, "description": "Glob patterns"},
            "limit_per_pattern": {"type": "number", "description": "Max files per pattern (default 20)"},
            "path": {"type": "string", "description": "Directory to search in"},
        },
        "required": ["patterns"],
    },
}

FIND_IN_FILES_TOOL: Dict[str, Any] = {
    "name": "FindInFiles",
    "description": "Search a regex across an explicit list of files; returns per-file matches with line numbers.",
    "input_schema": {
        "type": "object",
        "properties": {
            "files": {"type": "array", "items": {"type": "string"}, "description": "Files to scan"},
            "pattern": {"type": "string", "description": "Regular expression"},
            "case_sensitive": {"type": "boolean", "description": "Case-sensitive search (default false)"},
        },
        "required": ["files", "pattern"],
    },
}

SMART_SEARCH_TOOL: Dict[str, Any] = {
    "name": "SmartSearch",
    "description": (
        "Code-aware search from a natural query: finds definitions of "
        "functions/classes matching the query (e.g. 'def parse_args', "
        "'class Config') plus plain matches, across common languages."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "query": {"type": "string", "description": "Natural-language or code query"},
            "context": {"type": "string", "description": "Extra context words to rank results by (optional)"},
            "path": {"type": "string", "description": "Directory to search in"},
            "language": {"type": "string", "description": "Restrict to a language (python, js, ...)"},
        },
        "required": ["query"],
    },
}

REPO_MAP_TOOL: Dict[str, Any] = {
    "name": "RepoMap",
    "description": (
        "Generate a compact map of a repository under a token budget: the "
        "most important files with their top-level symbols (classes and "
        "functions), ranked by cross-file references."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "path": {"type": "string", "description": "Repository root (default: cwd)"},
            "token_budget": {"type": "integer", "description": "Approximate output budget in tokens"},
            "exclude_patterns": {"type": "array", "items": {"type": "string"}, "description": "Glob patterns to skip"},
        },
        "required": [],
    },
}

REPO_SUMMARY_TOOL: Dict[str, Any] = {
    "name": "RepoSummary",
    "description": "Summarize a repository: file counts by language, largest files, top-level layout.",
    "input_schema": {
        "type": "object",
        "properties": {
            "path": {"type": "string", "description": "Repository root (default: cwd)"},
            "max_tokens": {"type": "integer", "description": "Compact the output further"},
            "exclude_patterns": {"type": "array", "items": {"type": "string"}, "description": "Glob patterns to skip"},
        },
        "required": [],
    },
}

REPO_DEPS_TOOL: Dict[str, Any] = {
    "name": "RepoDependencies",   # reference name (definitions.py RepoDependencies)
    "description": "Extract module-level dependencies (imports) of a repository, aggregated per file.",
    "input_schema": {
        "type": "object",
        "properties": {
            "path": {"type": "string", "description": "Repository root (default: cwd)"},
            "module": {"type": "string", "description": "Only report dependencies of this module/file"},
            "depth": {"type": "integer", "description": "Transitive depth when module is given (default 1)"},
        },
        "required": [],
    },
}

SHELL_TOOL: Dict[str, Any] = {
    "name": "Shell",
    "description": (
        "Run a shell command. Commands are checked against an allowlist of "
        "known-safe programs and a denylist of destructive patterns. Output "
        "is truncated to 50 kB. Supports background mode with a timeout."
    ),
    "input_schema": {
        "type": "object",
        "properties": {
            "command": {"type": "string", "description": "Command line to execute"},
            "timeout": {"type": "number", "description": "Seconds before the command is killed (default 60)"},
            "background": {"type": "boolean", "description": "Run detached and return immediately"},
            "working_dir": {"type": "string", "description": "Directory to run in"},
            "current_dir": {"type": "string", "description": "Alias of working_dir (reference name)"},
        },
        "required": ["command"],
    },
}

BRAVE_SEARCH_TOOL: Dict[str, Any] = {
    "name": "brave_web_search",
    "description": "Web search via the Brave Search API (requires network + API key; unavailable offline).",
    "input_schema": {
        "type": "object",
        "properties": {
            "query": {"type": "string", "description": "Search query"},
            "count": {"type": "integer", "description": "Number of results (max 20)"},
            "offset": {"type": "integer", "description": "Pagination offset"},
        },
        "required": ["query"],
    },
}

TOOL_DEFINITIONS: List[Dict[str, Any]] = [
    GLOB_TOOL,
    GREP_TOOL,
    VIEW_TOOL,
    EDIT_TOOL,
    REPLACE_TOOL,
    LS_TOOL,
    REGEX_EDIT_TOOL,
    BATCH_GLOB_TOOL,
    FIND_IN_FILES_TOOL,
    SMART_SEARCH_TOOL,
    REPO_MAP_TOOL,
    REPO_SUMMARY_TOOL,
    REPO_DEPS_TOOL,
    SHELL_TOOL,
]

# The reference's "Anthropic variant" additionally exposes web search
# (fei/tools/definitions.py:407-441).
TOOL_DEFINITIONS_WITH_SEARCH: List[Dict[str, Any]] = TOOL_DEFINITIONS + [BRAVE_SEARCH_TOOL]
