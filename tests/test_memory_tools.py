"""Memory tool layer tests (direct-FS handlers + MemoryManager facade)."""

import pytest

from fei_amd.tools.memory_tools import MemoryManager, MemoryTools, create_memory_tools
from fei_amd.tools.registry import ToolRegistry


@pytest.fixture
def tools(memdir_base):
    return MemoryTools(base=memdir_base)


def test_create_view_list_delete(tools):
    out = tools.create({"subject": "remember this", "tags": "a,b",
                        "priority": "high", "body": "the body"})
    assert out["success"]
    mid = out["memory_id"]

    mem = tools.view({"memory_id": mid})
    assert mem["headers"]["Subject"] == "remember this"
    assert mem["content"] == "the body"

    lst = tools.list({})
    assert lst["count"] == 1
    assert lst["memories"][0]["memory_id"] == mid

    out = tools.delete({"memory_id": mid})
    assert out["success"] and out["folder"] == ".Trash"
    assert tools.list({})["count"] == 0


def test_search_and_tag(tools):
    tools.create({"subject": "gpu kernel notes", "tags": "gpu",
                  "body": "mfma tiling"})
    tools.create({"subject": "grocery list", "tags": "life"})
    out = tools.search({"query": "mfma", "with_content": True})
    assert out["count"] == 1
    out = tools.search_by_tag({"tag": "gpu"})
    assert out["count"] == 1


def test_registered_tool_names(memdir_base):
    reg = ToolRegistry()
    create_memory_tools(reg, base=memdir_base)
    names = reg.list_tools()
    for n in ["memory_search", "memory_semantic_search", "memory_index_build",
              "memory_create", "memory_view", "memory_list", "memory_delete",
              "memory_search_by_tag", "memdir_server_start",
              "memdir_server_stop", "memdir_server_status"]:
        assert n in names
    out = reg.execute_tool("memory_create", {"subject": "via registry"})
    assert out["success"]
    out = reg.execute_tool("memory_search", {"query": "registry"})
    assert out["count"] == 1


def test_semantic_index_cpu(tools):
    """CPU path: build a tiny index with the bge encoder on CPU and query it
    (GPU speed comes from the same code path on MFMA)."""
    tools.create({"subject": "rocprof kernel profiling", "tags": "gpu",
                  "body": "per-kernel time breakdown on MI355X"})
    tools.create({"subject": "pasta recipe", "tags": "cooking",
                  "body": "boil water, add salt"})
    out = tools.index_build({})
    assert out["indexed"] == 2
    res = tools.semantic_search({"query": "GPU profiling", "topk": 2})
    assert res["count"] == 2
    assert all("score" in m for m in res["results"])


def test_memory_manager_save_conversation(memdir_base):
    mgr = MemoryManager(base=memdir_base)
    messages = [
        {"role": "user", "content": "how do I profile kernels?"},
        {"role": "assistant", "content": [
            {"type": "text", "text": "use rocprofv3"},
            {"type": "tool_use", "name": "Shell", "id": "1", "input": {}},
        ]},
    ]
    out = mgr.save_conversation(messages, subject="profiling chat")
    assert out["success"]
    results = mgr.recall("rocprofv3")
    assert len(results) == 1
    assert "use rocprofv3" in results[0]["content"]


def test_embedding_index_persistence(memdir_base):
    """save() -> fresh instance load() -> identical search results."""
    from fei_amd.memdir.embed_index import EmbeddingIndex
    tools = MemoryTools(base=memdir_base)
    tools.create({"subject": "hip kernels", "body": "mfma and lds tiling"})
    tools.create({"subject": "lunch plan", "body": "soup and bread"})

    idx = EmbeddingIndex(base=memdir_base)
    n = idx.build()
    assert n == 2
    want = idx.search("matrix core tiling", topk=2)
    idx.save()

    idx2 = EmbeddingIndex(base=memdir_base)
    assert idx2.load()
    assert idx2.ids == idx.ids
    got = idx2.search("matrix core tiling", topk=2)
    # fp16 round trip through the .npy file: same ranking, close scores
    assert [i for i, _ in got] == [i for i, _ in want]
    for (_, a), (_, b) in zip(got, want):
        assert abs(a - b) < 1e-2


def test_embedding_index_add_texts(memdir_base):
    from fei_amd.memdir.embed_index import EmbeddingIndex
    idx = EmbeddingIndex(base=memdir_base)
    idx.add_texts(["alpha doc about gpus"], ["id1"])
    idx.add_texts(["beta doc about cooking"], ["id2"])
    assert idx.embeddings.shape[0] == 2 and idx.ids == ["id1", "id2"]
    # the encoder is random-init: only an identical text guarantees rank 1
    res = idx.search("alpha doc about gpus", topk=2)
    assert res[0][0] == "id1"
    assert res[0][1] > 0.999


def test_reference_field_names(tools):
    """Reference argument names: content (=body), limit, status."""
    out = tools.create({"subject": "ref style", "content": "via content field"})
    assert out["success"]
    got = tools.view({"memory_id": out["memory_id"]})
    assert got["content"] == "via content field"
    for i in range(5):
        tools.create({"subject": f"bulk {i}", "tags": "bulk"})
    assert tools.list({"limit": 2})["count"] >= 2
    assert len(tools.list({"limit": 2})["memories"]) == 2
    res = tools.search({"query": "bulk", "limit": 3})
    assert len(res["results"]) == 3


def test_live_index_updated_on_create(tools):
    tools.create({"subject": "seed memory", "body": "initial content"})
    tools.index_build({})
    n0 = tools.index().embeddings.shape[0]
    tools.create({"subject": "fresh gpu insight",
                  "body": "the new memory must be searchable immediately"})
    assert tools.index().embeddings.shape[0] == n0 + 1
    res = tools.semantic_search(
        {"query": "fresh gpu insight\n\nthe new memory must be searchable immediately",
         "topk": 1})
    assert res["results"][0]["headers"]["Subject"] == "fresh gpu insight"


def test_live_index_removal_on_delete(tools):
    a = tools.create({"subject": "kept entry", "body": "stay"})
    b = tools.create({"subject": "doomed entry", "body": "go away"})
    tools.index_build({})
    n0 = tools.index().embeddings.shape[0]
    tools.delete({"memory_id": b["memory_id"]})
    assert tools.index().embeddings.shape[0] == n0 - 1
    res = tools.semantic_search({"query": "anything", "topk": 10})
    subs = [m["headers"]["Subject"] for m in res["results"]]
    assert "doomed entry" not in subs and "kept entry" in subs


def test_fts_keyword_search(tools):
    tools.create({"subject": "rocprof profiling guide", "tags": "gpu",
                  "body": "collect kernel traces with rocprofv3 on MI355X"})
    tools.create({"subject": "pasta recipe", "tags": "food",
                  "body": "boil water and add the rocprof... no, salt"})
    tools.create({"subject": "unrelated", "body": "nothing to see"})
    out = tools.index_build({})
    assert out["fts_indexed"] == 3
    res = tools.keyword_search({"query": "rocprof kernel traces"})
    assert res["count"] >= 1
    assert res["results"][0]["headers"]["Subject"] == "rocprof profiling guide"

    # live updates: create is searchable, delete disappears
    made = tools.create({"subject": "fresh fts entry",
                         "body": "immediately findable zebra"})
    res = tools.keyword_search({"query": "zebra"})
    assert res["count"] == 1
    tools.delete({"memory_id": made["memory_id"]})
    assert tools.keyword_search({"query": "zebra"})["count"] == 0

    # operator/punctuation injection cannot break the query
    assert tools.keyword_search({"query": 'NEAR( "unbalanced OR *'})["count"] >= 0
