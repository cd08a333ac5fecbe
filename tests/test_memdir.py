"""memdir format + primitives tests (format contract: SURVEY.md §2.3)."""

import os
import re

from fei_amd.memdir import utils as mu
from fei_amd.memdir.folders import MemdirFolderManager
from fei_amd.memdir.filter import FilterManager, MemoryFilter
from fei_amd.memdir.archiver import MemoryArchiver


FILENAME_RE = re.compile(r"^\d+\.[0-9a-f]{8}\.[^:]+:2,[A-Z]*$")


def test_filename_grammar(memdir_base):
    name = mu.create_memory(base=memdir_base, headers={"Subject": "s"}, body="b",
                            flags="FS")
    assert FILENAME_RE.match(name)
    meta = mu.parse_memory_filename(name)
    assert meta["flags"] == ["F", "S"]  # sorted, deduped
    assert isinstance(meta["timestamp"], int)


def test_content_roundtrip(memdir_base):
    headers = {"Subject": "Test memo", "Tags": "a,b"}
    body = "line one\nline two\n---inline dashes ok\nend"
    name = mu.create_memory(base=memdir_base, headers=headers, body=body)
    mems = mu.list_memories("", "new", include_content=True, base=memdir_base)
    assert len(mems) == 1
    assert mems[0]["headers"]["Subject"] == "Test memo"
    assert mems[0]["headers"]["Tags"] == "a,b"
    assert mems[0]["content"] == body
    assert "Date" in mems[0]["headers"]


def test_atomic_write_leaves_no_tmp(memdir_base):
    mu.create_memory(base=memdir_base, headers={"Subject": "x"}, body="y")
    assert os.listdir(os.path.join(memdir_base, "tmp")) == []
    assert len(os.listdir(os.path.join(memdir_base, "new"))) == 1


def test_move_and_reflag(memdir_base):
    name = mu.create_memory(base=memdir_base, headers={"Subject": "m"}, body="")
    assert mu.move_memory(name, "", ".Projects", src_status="new",
                          dst_status="cur", base=memdir_base)
    mems = mu.list_memories(".Projects", "cur", base=memdir_base)
    assert len(mems) == 1
    new_name = mu.update_memory_flags(mems[0]["filename"], ".Projects", "cur",
                                      "SF", base=memdir_base)
    assert new_name.endswith(":2,FS")


def test_find_memory_by_unique(memdir_base):
    name = mu.create_memory(base=memdir_base, headers={"Subject": "f"}, body="")
    unique = mu.parse_memory_filename(name)["unique"]
    loc = mu.find_memory(unique, base=memdir_base)
    assert loc == ("", "new", name)
    mem = mu.read_memory(*loc, base=memdir_base)
    assert mem["headers"]["Subject"] == "f"


def test_folder_manager(memdir_base):
    mgr = MemdirFolderManager(memdir_base)
    assert mgr.create_folder(".Projects/python")
    assert mgr.folder_exists(".Projects/python")
    mu.create_memory(".Projects/python", {"Subject": "p", "Tags": "code"},
                     "body", flags="F", base=memdir_base, status="cur")
    stats = mgr.get_folder_stats(".Projects/python")
    assert stats["total"] == 1
    assert stats["flags"] == {"F": 1}
    assert stats["tags"] == {"code": 1}
    assert mgr.rename_folder(".Projects/python", ".Projects/py")
    assert not mgr.folder_exists(".Projects/python")


def test_delete_folder_evacuates_to_trash(memdir_base):
    mgr = MemdirFolderManager(memdir_base)
    mgr.create_folder(".Temp")
    mu.create_memory(".Temp", {"Subject": "keepme"}, "", base=memdir_base, status="cur")
    assert mgr.delete_folder(".Temp")
    trash = mu.list_memories(".Trash", "cur", include_content=True, base=memdir_base)
    assert len(trash) == 1
    assert trash[0]["headers"]["Subject"] == "keepme"


def test_bulk_tag(memdir_base):
    mgr = MemdirFolderManager(memdir_base)
    mu.create_memory("", {"Subject": "a", "Tags": "x"}, "", base=memdir_base, status="cur")
    mu.create_memory("", {"Subject": "b"}, "", base=memdir_base, status="cur")
    n = mgr.bulk_tag_folder("", ["y"], mode="add")
    assert n == 2
    mems = mu.list_memories("", "cur", include_content=True, base=memdir_base)
    tags = sorted(m["headers"]["Tags"] for m in mems)
    assert tags == ["x,y", "y"]


def test_filters_move_and_graduate(memdir_base):
    mu.ensure_folder("", memdir_base)
    mu.create_memory("", {"Subject": "todo: fix the bug"}, "", base=memdir_base)
    mu.create_memory("", {"Subject": "plain note"}, "", base=memdir_base)
    mgr = FilterManager(memdir_base)
    report = mgr.process_memories("")
    assert report["processed"] == 2
    # todo memory moved to .ToDoLater, plain note graduated new -> cur
    assert len(mu.list_memories(".ToDoLater", "cur", base=memdir_base)) == 1
    assert len(mu.list_memories("", "cur", base=memdir_base)) == 1
    assert len(mu.list_memories("", "new", base=memdir_base)) == 0


def test_custom_filter_flags(memdir_base):
    mu.create_memory("", {"Subject": "urgent thing", "Priority": "high"}, "",
                     base=memdir_base)
    f = MemoryFilter("prio", [{"field": "Priority", "pattern": "high"}],
                     [{"action": "flag", "flags": "FP"}])
    FilterManager(memdir_base, [f]).process_memories("", move_unmatched_to_cur=False)
    mems = mu.list_memories("", "new", base=memdir_base)
    assert sorted(mems[0]["metadata"]["flags"]) == ["F", "P"]


def test_archiver_archives_old(memdir_base):
    mu.ensure_folder("", memdir_base)
    old_name = mu.generate_filename("")
    # fabricate an old memory (200 days old)
    import time
    ts = int(time.time() - 200 * 86400)
    old_name = f"{ts}.{'a'*8}.host:2,"
    path = os.path.join(mu.ensure_folder("", memdir_base), "cur", old_name)
    with open(path, "w") as f:
        f.write("Subject: old\n---\nbody")
    mu.create_memory("", {"Subject": "fresh"}, "", base=memdir_base, status="cur")
    arch = MemoryArchiver(memdir_base)
    n = arch.archive_old_memories(age_days=90)
    assert n == 1
    year = time.gmtime(ts).tm_year
    assert len(mu.list_memories(f".Archive/{year}", "cur", base=memdir_base)) == 1
    assert len(mu.list_memories("", "cur", base=memdir_base)) == 1


def test_archiver_importance_eviction(memdir_base):
    mu.ensure_folder("", memdir_base)
    for i in range(5):
        mu.create_memory("", {"Subject": f"m{i}"}, "", base=memdir_base,
                         status="cur", flags="F" if i < 2 else "")
    arch = MemoryArchiver(memdir_base)
    evicted = arch.apply_retention_policies(max_per_folder=2)
    assert evicted == 3
    kept = mu.list_memories("", "cur", base=memdir_base)
    assert all("F" in m["metadata"]["flags"] for m in kept)


def test_create_samples(memdir_base):
    from fei_amd.memdir.create_samples import create_samples
    n = create_samples(count=10, base=memdir_base)
    assert n == 10
    total = sum(len(mu.list_memories(f, s, base=memdir_base))
                for f in mu.list_folders(memdir_base) for s in ("cur", "new"))
    assert total == 10


def test_archiver_cleanup_rules_and_trash_expiry(memdir_base):
    import os, time
    mu.ensure_folder(".ToDoLater", memdir_base)
    mu.ensure_folder(".Trash", memdir_base)
    old_ts = int(time.time() - 400 * 86400)
    for folder in (".ToDoLater", ".Trash"):
        path = os.path.join(mu.get_memdir_base(memdir_base), folder, "cur",
                            f"{old_ts}.{'b'*8}.host:2,")
        with open(path, "w") as f:
            f.write("Subject: stale\n---\n")
    arch = MemoryArchiver(memdir_base)
    assert arch.cleanup_memories() == 1          # ToDoLater -> Trash
    assert arch.empty_trash() >= 1               # stale trash deleted


def test_memory_content_without_separator():
    headers, body = mu.parse_memory_content("just a plain body line")
    assert headers == {}
    assert body == "just a plain body line"


def test_move_memory_missing_file(memdir_base):
    mu.ensure_folder("", memdir_base)
    assert mu.move_memory("nope", "", ".Trash", base=memdir_base) is False


def test_fts_index_edges(memdir_base):
    from fei_amd.memdir.fts_index import FtsIndex
    idx = FtsIndex(base=memdir_base)
    assert idx.build() == 0
    assert idx.search("anything") == []          # empty index
    assert idx.search("") == []                  # empty query
    idx.add("", "cur", "f1:2,", "subj one", "t", "alpha beta")
    idx.add("", "cur", "f2:2,", "subj two", "t", "beta gamma")
    assert idx.count() == 2
    keys = [k for k, _ in idx.search("beta")]
    assert len(keys) == 2
    assert idx.remove("f1") == 1
    assert idx.count() == 1
    # persistence across instances (same sqlite file)
    idx.close()
    idx2 = FtsIndex(base=memdir_base)
    assert idx2.count() == 1
