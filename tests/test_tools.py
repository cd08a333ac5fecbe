"""Tool layer tests (parity with reference fei/tests/test_tools.py:21-241:
real tmp dirs, assert counts/contents)."""

import os

import pytest

from fei_amd.tools.code import (
    CodeEditor, DirectoryExplorer, FileViewer, GlobFinder, GrepTool, ShellRunner,
)


@pytest.fixture
def workspace(tmp_path):
    (tmp_path / "src").mkdir()
    (tmp_path / "src" / "main.py").write_text("def main():\n    return 42\n")
    (tmp_path / "src" / "util.py").write_text("def helper(x):\n    return x + 1\n\nVALUE = 3\n")
    (tmp_path / "README.md").write_text("# Project\nhello world\n")
    (tmp_path / "data.bin").write_bytes(b"\x00\x01\x02binary")
    return tmp_path


def test_glob_finds_python(workspace):
    finder = GlobFinder()
    files = finder.find("**/*.py", str(workspace))
    assert len(files) == 2
    assert all(f.endswith(".py") for f in files)


def test_glob_jail(tmp_path):
    finder = GlobFinder(base_path=str(tmp_path))
    with pytest.raises(PermissionError):
        finder.find("*", "/etc")


def test_glob_batch(workspace):
    finder = GlobFinder()
    out = finder.batch(["**/*.py", "*.md"], str(workspace))
    assert len(out["**/*.py"]) == 2
    assert len(out["*.md"]) == 1


def test_grep_matches(workspace):
    grep = GrepTool()
    matches = grep.search(r"def \w+", str(workspace))
    assert len(matches) == 2
    assert {m["line"] for m in matches} == {1}


def test_grep_include_filter(workspace):
    grep = GrepTool()
    matches = grep.search("hello", str(workspace), include="*.py")
    assert matches == []
    matches = grep.search("hello", str(workspace), include="*.md")
    assert len(matches) == 1


def test_grep_skips_binary(workspace):
    grep = GrepTool()
    matches = grep.search("binary", str(workspace))
    assert all("data.bin" not in m["file"] for m in matches)


def test_find_in_files(workspace):
    grep = GrepTool()
    f = str(workspace / "src" / "util.py")
    out = grep.find_in_files([f, str(workspace / "missing.py")], "VALUE")
    assert out[f][0]["line"] == 4
    assert "error" in out[str(workspace / "missing.py")][0]


def test_edit_unique_match(workspace):
    editor = CodeEditor()
    f = str(workspace / "src" / "main.py")
    result = editor.edit_file(f, "return 42", "return 43")
    assert result.get("success")
    assert "return 43" in open(f).read()
    # backup was created
    backups = os.listdir(str(workspace / "src" / ".fei_backups"))
    assert len(backups) == 1


def test_edit_rejects_ambiguous(tmp_path):
    f = tmp_path / "x.py"
    f.write_text("a = 1\na = 1\n")
    result = CodeEditor().edit_file(str(f), "a = 1", "a = 2")
    assert "error" in result and "2 times" in result["error"]


def test_edit_empty_old_string_creates(tmp_path):
    """Defect-fix check: empty old_string creates the file (the reference
    promised this but couldn't do it — SURVEY.md defect list)."""
    f = tmp_path / "new_file.py"
    result = CodeEditor().edit_file(str(f), "", "x = 1\n")
    assert result.get("success")
    assert f.read_text() == "x = 1\n"


def test_regex_edit_validates_python(tmp_path):
    f = tmp_path / "x.py"
    f.write_text("def f():\n    return 1\n")
    # breaking the syntax must be rolled back
    result = CodeEditor().regex_replace(str(f), r"def f\(\):", "def f(:")
    assert "error" in result
    assert f.read_text() == "def f():\n    return 1\n"
    # a good edit goes through
    result = CodeEditor().regex_replace(str(f), r"return 1", "return 2")
    assert result.get("success") and result["replacements"] == 1


def test_view_offset_limit(workspace):
    viewer = FileViewer()
    f = str(workspace / "src" / "util.py")
    out = viewer.view(f, offset=2, limit=1)
    assert out["lines_shown"] == 1
    assert out["content"].startswith("2\t")
    assert out["total_lines"] == 4


def test_view_binary_rejected(workspace):
    out = FileViewer().view(str(workspace / "data.bin"))
    assert "error" in out


def test_ls(workspace):
    out = DirectoryExplorer().list_directory(str(workspace))
    names = [e["path"] for e in out["entries"]]
    assert "src/" in names and "README.md" in names


def test_ls_ignore(workspace):
    out = DirectoryExplorer().list_directory(str(workspace), ignore=["*.md", "*.bin"])
    names = [e["path"] for e in out["entries"]]
    assert "README.md" not in names


def test_shell_allowlist():
    runner = ShellRunner()
    out = runner.run("echo hello")
    assert out["success"] and out["stdout"].strip() == "hello"
    out = runner.run("rm -rf /")
    assert "error" in out
    out = runner.run("nc -l 1234")
    assert "error" in out and "allowlist" in out["error"]


def test_shell_denylist_patterns():
    runner = ShellRunner()
    assert "error" in runner.run("shutdown now")
    assert "error" in runner.run("vi file.txt")


def test_shell_pipeline_heads_checked():
    runner = ShellRunner()
    out = runner.run("echo hi | wc -l")
    assert out["success"]
    out = runner.run("echo hi | nc example.com 80")
    assert "error" in out


def test_shell_timeout():
    out = ShellRunner().run("sleep 5", timeout=0.3)
    assert "error" in out and "timed out" in out["error"]


def test_shell_background_mode(tmp_path):
    runner = ShellRunner()
    marker = tmp_path / "done.txt"
    out = runner.run(f"sh -c 'sleep 0.2; touch {marker}'", timeout=10,
                     background=True)
    assert out.get("background") and out.get("pid")
    import time
    deadline = time.time() + 5
    while not marker.exists() and time.time() < deadline:
        time.sleep(0.05)
    assert marker.exists()


# -- multi-language post-edit validators (reference code.py:827-932) ---------

def test_validator_json_rejects_bad_edit(tmp_path):
    from fei_amd.tools.code import CodeEditor

    ed = CodeEditor()
    p = tmp_path / "cfg.json"
    p.write_text('{"a": 1}')
    out = ed.regex_replace(str(p), r"1", "1,,")       # invalid json
    assert "error" in out and "json" in out["error"]
    assert p.read_text() == '{"a": 1}'                # unchanged
    ok = ed.regex_replace(str(p), r"1", "2")
    assert ok.get("success")


def test_validator_yaml_rejects_bad_edit(tmp_path):
    from fei_amd.tools.code import CodeEditor

    ed = CodeEditor()
    p = tmp_path / "c.yaml"
    p.write_text("key: value\n")
    out = ed.regex_replace(str(p), "key: value", "key: [unclosed")
    assert "error" in out and "yaml" in out["error"]


def test_validator_python_still_guards(tmp_path):
    from fei_amd.tools.code import CodeEditor

    ed = CodeEditor()
    p = tmp_path / "m.py"
    p.write_text("def f():\n    return 1\n")
    out = ed.regex_replace(str(p), "return 1", "return (")
    assert "error" in out and "syntax" in out["error"]


def test_validator_optional_absent_skips(tmp_path):
    """Languages whose optional checker (esprima/tree-sitter) is not
    importable in this image pass through ungated — graceful absence."""
    from fei_amd.tools.code import CodeEditor

    ed = CodeEditor()
    p = tmp_path / "app.js"
    p.write_text("function f() { return 1; }\n")
    out = ed.regex_replace(str(p), "return 1", "return (((")
    assert out.get("success")        # no JS checker available here


def test_validator_treesitter_gate(tmp_path, monkeypatch):
    """With a (fake) tree-sitter present, a parse-error tree blocks the
    edit."""
    import sys

    from fei_amd.tools import repomap as rm
    from fei_amd.tools.code import CodeEditor

    class _Node:
        has_error = True

    class _Tree:
        root_node = _Node()

    class _Parser:
        def parse(self, data):
            return _Tree()

    class _Lang:
        def query(self, q):
            return object()

    class _TSL:
        def get_parser(self, lang):
            return _Parser()

        def get_language(self, lang):
            return _Lang()

    monkeypatch.setitem(sys.modules, "tree_sitter_languages", _TSL())
    monkeypatch.setattr(rm, "_TS_CACHE", {})
    ed = CodeEditor()
    p = tmp_path / "lib.rs"
    p.write_text("fn main() {}\n")
    out = ed.regex_replace(str(p), r"\{\}", "{ let x = ; }")
    assert "error" in out and "rust" in out["error"]


def test_validator_explicit_list_overrides(tmp_path):
    from fei_amd.tools.code import CodeEditor

    ed = CodeEditor()
    p = tmp_path / "m.py"
    p.write_text("x = 1\n")
    # empty list disables validation entirely (reference semantics)
    out = ed.regex_replace(str(p), "x = 1", "x = (", validators=[])
    assert out.get("success")
