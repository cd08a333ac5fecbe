"""memdir query language tests (grammar: SURVEY.md §2.3 / search.py parity)."""

import time

from fei_amd.memdir import utils as mu
from fei_amd.memdir.search import (
    SearchQuery, format_results, parse_search_args, search, search_memories,
)


def _seed(memdir_base):
    mu.create_memory("", {"Subject": "GPU kernel notes", "Tags": "gpu,perf",
                          "Priority": "high"},
                     "rocprof shows 85% HBM utilization", flags="F",
                     base=memdir_base, status="cur")
    mu.create_memory("", {"Subject": "Shopping list", "Tags": "personal"},
                     "milk and eggs", base=memdir_base, status="cur")
    mu.create_memory(".Projects", {"Subject": "Agent loop design", "Tags": "agent"},
                     "one tool round per chat turn", base=memdir_base, status="cur")


def test_keyword_or_semantics(memdir_base):
    _seed(memdir_base)
    res = search("kernel", base=memdir_base)
    assert len(res) == 1
    res = search("kernel milk", base=memdir_base)  # OR across keywords
    assert len(res) == 2


def test_field_contains_and_equals(memdir_base):
    _seed(memdir_base)
    assert len(search("Subject:gpu", base=memdir_base)) == 1
    assert len(search("Priority=high", base=memdir_base)) == 1
    assert len(search("Priority!=high", base=memdir_base)) == 2


def test_tag_and_flag_shortcuts(memdir_base):
    _seed(memdir_base)
    assert len(search("#gpu", base=memdir_base)) == 1
    assert len(search("#agent", base=memdir_base)) == 1
    assert len(search("+F", base=memdir_base)) == 1


def test_regex_query(memdir_base):
    _seed(memdir_base)
    res = search("/HBM util/", base=memdir_base)
    assert len(res) == 1


def test_relative_date(memdir_base):
    _seed(memdir_base)
    assert len(search("date>now-1d", base=memdir_base)) == 3
    assert len(search("date<now-1d", base=memdir_base)) == 0


def test_sort_limit_offset(memdir_base):
    for i in range(5):
        mu.create_memory("", {"Subject": f"note {i}"}, "", base=memdir_base,
                         status="cur")
    q = parse_search_args("note sort:Subject limit:2")
    q.sort_reverse = False
    res = search_memories(q, base=memdir_base)
    assert [m["headers"]["Subject"] for m in res] == ["note 0", "note 1"]
    q = parse_search_args("note limit:2 offset:2")
    q.sort_field, q.sort_reverse = "Subject", False
    res = search_memories(q, base=memdir_base)
    assert [m["headers"]["Subject"] for m in res] == ["note 2", "note 3"]


def test_with_content_modifier(memdir_base):
    _seed(memdir_base)
    res = search("kernel", base=memdir_base)
    assert "content" not in res[0]
    res = search("kernel with_content", base=memdir_base)
    assert "rocprof" in res[0]["content"]


def test_folder_restriction(memdir_base):
    _seed(memdir_base)
    res = search("folder:.Projects tool", base=memdir_base)
    assert len(res) == 1
    assert res[0]["folder"] == ".Projects"


def test_status_header_vs_maildir_status(memdir_base):
    mu.create_memory("", {"Subject": "s", "Status": "in-progress"}, "",
                     base=memdir_base, status="cur")
    # "Status" header wins for field lookups; maildir status via status: filter
    res = search("Status:in-progress", base=memdir_base)
    assert len(res) == 1
    q = SearchQuery().set_statuses(["new"])
    assert search_memories(q, base=memdir_base) == []


def test_quoted_phrase(memdir_base):
    _seed(memdir_base)
    res = search('"tool round"', base=memdir_base)
    assert len(res) == 1


def test_format_results(memdir_base):
    _seed(memdir_base)
    res = search("kernel with_content", base=memdir_base)
    for fmt in ("text", "json", "csv", "compact"):
        out = format_results(res, fmt)
        assert "GPU kernel notes" in out or "GPU kernel notes" in out
