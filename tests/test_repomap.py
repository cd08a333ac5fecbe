"""RepoMapper tests."""

from fei_amd.tools.repomap import RepoMapper


def _repo(tmp_path):
    (tmp_path / "pkg").mkdir()
    (tmp_path / "pkg" / "core.py").write_text(
        "class Engine:\n    def run(self):\n        pass\n\ndef helper_function():\n    pass\n")
    (tmp_path / "pkg" / "app.py").write_text(
        "from pkg.core import Engine\n\ndef main():\n    e = Engine()\n    helper_function()\n")
    (tmp_path / "native.hip").write_text(
        "__global__ void rmsnorm_kernel(float* x) {}\n")
    return tmp_path


def test_symbols_python_ast(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    syms = mapper.symbols()
    core = [s for p, s in syms.items() if p.endswith("core.py")][0]
    names = {n for n, _ in core}
    assert {"Engine", "Engine.run", "helper_function"} <= names


def test_symbols_hip_kernel(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    syms = mapper.symbols()
    hip = [s for p, s in syms.items() if p.endswith("native.hip")][0]
    assert ("rmsnorm_kernel", "kernel") in hip


def test_map_ranks_referenced_files(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    out = mapper.generate_map(token_budget=500)
    # core.py is referenced by app.py, so it should appear first
    assert out.index("core.py") < out.index("app.py")
    assert "class Engine" in out


def test_map_budget_respected(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    out = mapper.generate_map(token_budget=10)
    assert len(out) <= 10 * 4 + 80


def test_dependencies(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    deps = mapper.dependencies()
    assert deps["pkg/app.py"] == ["pkg.core"]


def test_summary(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    s = mapper.summary()
    assert s["file_count"] == 3
    assert s["by_extension"][".py"] == 2
    assert "pkg" in s["top_level_dirs"]


def test_json_output(tmp_path):
    mapper = RepoMapper(str(_repo(tmp_path)))
    j = mapper.generate_json()
    assert "pkg/core.py" in j["files"]


def test_excluded_dirs_skipped(tmp_path):
    (tmp_path / "node_modules").mkdir()
    (tmp_path / "node_modules" / "x.py").write_text("def hidden(): pass\n")
    (tmp_path / "ok.py").write_text("def visible(): pass\n")
    files = RepoMapper(str(tmp_path)).source_files()
    assert len(files) == 1 and files[0].endswith("ok.py")


def test_budget_zero_still_returns_something(tmp_path):
    (tmp_path / "a.py").write_text("def f(): pass\n")
    out = RepoMapper(str(tmp_path)).generate_map(token_budget=0)
    assert isinstance(out, str)
