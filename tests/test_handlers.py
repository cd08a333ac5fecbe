"""Handler-level tests (SmartSearch language regexes, shell handler)."""

from fei_amd.tools import handlers as h


def test_smart_search_finds_definition(tmp_path):
    (tmp_path / "m.py").write_text("def parse_args(argv):\n    return argv\n")
    (tmp_path / "m.js").write_text("function parseArgs(x) { return x; }\n")
    out = h.smart_search_handler({"query": "def parse_args", "path": str(tmp_path)})
    assert out["identifier"] == "parse_args"
    assert any(d["file"].endswith("m.py") for d in out["definitions"])


def test_smart_search_language_filter(tmp_path):
    (tmp_path / "m.py").write_text("class Config:\n    pass\n")
    (tmp_path / "m.rs").write_text("struct Config {}\n")
    out = h.smart_search_handler({"query": "Config", "path": str(tmp_path),
                                  "language": "rust"})
    assert all(d["language"] == "rust" for d in out["definitions"])


def test_shell_handler(tmp_path):
    out = h.shell_handler({"command": "pwd", "working_dir": str(tmp_path)})
    assert out["success"]
    assert out["stdout"].strip() == str(tmp_path)


def test_batch_glob_handler(tmp_path):
    (tmp_path / "a.py").write_text("")
    out = h.batch_glob_handler({"patterns": ["*.py", "*.md"], "path": str(tmp_path)})
    assert len(out["results"]["*.py"]) == 1
    assert out["results"]["*.md"] == []


def test_repo_handlers(tmp_path):
    (tmp_path / "x.py").write_text("import os\n\ndef f():\n    pass\n")
    out = h.repo_map_handler({"path": str(tmp_path)})
    assert "x.py" in out["map"]
    out = h.repo_summary_handler({"path": str(tmp_path)})
    assert out["file_count"] == 1
    out = h.repo_deps_handler({"path": str(tmp_path)})
    assert out["dependencies"]["x.py"] == ["os"]


def test_reference_schema_fields(tmp_path):
    """The reference's tool argument names must all work: case_sensitive,
    limit_per_pattern, validators, current_dir, depth, exclude_patterns."""
    from fei_amd.tools.handlers import (
        batch_glob_handler, find_in_files_handler, regex_edit_handler,
        repo_map_handler, shell_handler,
    )
    f = tmp_path / "a.py"
    f.write_text("VALUE = 1\nvalue = 2\n")
    (tmp_path / "b.txt").write_text("x")

    out = find_in_files_handler({"files": [str(f)], "pattern": "VALUE"})
    assert len(out["results"][str(f)]) == 2          # default: insensitive
    out = find_in_files_handler({"files": [str(f)], "pattern": "VALUE",
                                 "case_sensitive": True})
    assert len(out["results"][str(f)]) == 1

    out = batch_glob_handler({"patterns": ["*"], "path": str(tmp_path),
                              "limit_per_pattern": 1})
    assert len(out["results"]["*"]) == 1

    out = regex_edit_handler({"file_path": str(f), "pattern": "= 1",
                              "replacement": "= 10", "validators": ["ast"]})
    assert out["success"]
    out = regex_edit_handler({"file_path": str(f), "pattern": r"VALUE =",
                              "replacement": "VALUE ==",   # would break ast
                              "validators": []})           # explicitly off
    assert out.get("success")                              # not validated

    out = shell_handler({"command": "pwd", "current_dir": str(tmp_path)})
    assert str(tmp_path) in out.get("stdout", "")

    out = repo_map_handler({"path": str(tmp_path),
                            "exclude_patterns": ["*.py"]})
    assert "a.py" not in out["map"]
