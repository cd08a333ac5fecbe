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


# -- tree-sitter path (reference repomap.py:160-281) -------------------------

class _FakeNode:
    def __init__(self, name):
        self.text = name.encode()


class _FakeQuery:
    def __init__(self, captures):
        self._caps = captures

    def captures(self, root):
        return self._caps


class _FakeTSL:
    """Stands in for tree_sitter_languages (not installed in this image):
    exercises OUR glue — language selection, capture normalization, dedupe
    and the unavailable-grammar fallback."""
    def __init__(self, captures, known=("javascript", "cpp")):
        self._captures = captures
        self._known = known

    def get_parser(self, lang):
        if lang not in self._known:
            raise LookupError(lang)

        class P:
            def parse(self_inner, data):
                class T:
                    root_node = None
                return T()
        return P()

    def get_language(self, lang):
        if lang not in self._known:
            raise LookupError(lang)
        caps = self._captures

        class L:
            def query(self_inner, q):
                assert "@function" in q or "@class" in q or "@method" in q
                return _FakeQuery(caps)
        return L()


def test_treesitter_extraction_used_when_importable(tmp_path, monkeypatch):
    import sys

    from fei_amd.tools import repomap as rm

    fake = _FakeTSL([(_FakeNode("renderWidget"), "function"),
                     (_FakeNode("Widget"), "class"),
                     (_FakeNode("Widget"), "class")])      # dup: dedup check
    monkeypatch.setitem(sys.modules, "tree_sitter_languages", fake)
    monkeypatch.setattr(rm, "_TS_CACHE", {})
    js = tmp_path / "app.js"
    js.write_text("class Widget {}\nfunction renderWidget() {}\n")
    mapper = rm.RepoMapper(str(tmp_path))
    syms = mapper.extract_symbols(str(js))
    assert ("renderWidget", "function") in syms
    assert syms.count(("Widget", "class")) == 1


def test_treesitter_dict_captures_api(tmp_path, monkeypatch):
    """tree-sitter >= 0.22 returns {name: [nodes]} from captures()."""
    import sys

    from fei_amd.tools import repomap as rm

    fake = _FakeTSL({"class": [_FakeNode("Engine")],
                     "method": [_FakeNode("Engine::run")]})
    monkeypatch.setitem(sys.modules, "tree_sitter_languages", fake)
    monkeypatch.setattr(rm, "_TS_CACHE", {})
    cc = tmp_path / "engine.cpp"
    cc.write_text("class Engine {};\nvoid Engine::run() {}\n")
    syms = rm.RepoMapper(str(tmp_path)).extract_symbols(str(cc))
    assert ("Engine", "class") in syms
    assert ("Engine::run", "method") in syms


def test_treesitter_absent_falls_back_to_regex(tmp_path, monkeypatch):
    import sys

    from fei_amd.tools import repomap as rm

    monkeypatch.setitem(sys.modules, "tree_sitter_languages", None)
    monkeypatch.setattr(rm, "_TS_CACHE", {})
    js = tmp_path / "app.js"
    js.write_text("function renderWidget() {}\nclass Widget {}\n")
    syms = rm.RepoMapper(str(tmp_path)).extract_symbols(str(js))
    assert ("renderWidget", "function") in syms
    assert ("Widget", "class") in syms
    hip = tmp_path / "k.hip"
    hip.write_text("__global__ void k_decode(int* x) {}\n")
    syms = rm.RepoMapper(str(tmp_path)).extract_symbols(str(hip))
    assert ("k_decode", "kernel") in syms


def test_treesitter_unknown_grammar_falls_back(tmp_path, monkeypatch):
    import sys

    from fei_amd.tools import repomap as rm

    fake = _FakeTSL([], known=("javascript",))       # no 'rust' grammar
    monkeypatch.setitem(sys.modules, "tree_sitter_languages", fake)
    monkeypatch.setattr(rm, "_TS_CACHE", {})
    rs = tmp_path / "lib.rs"
    rs.write_text("pub struct Conn {}\nfn connect() {}\n")
    syms = rm.RepoMapper(str(tmp_path)).extract_symbols(str(rs))
    assert ("connect", "function") in syms
    assert ("Conn", "class") in syms
