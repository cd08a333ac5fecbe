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
