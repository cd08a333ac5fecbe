"""Concurrency tests the reference lacks (SURVEY §5 lists its known races):
parallel tool dispatch, parallel memdir writes, parallel chain appends."""

import threading

from fei_amd.memdir import utils as mu
from fei_amd.tools.registry import ToolRegistry


def test_registry_parallel_dispatch():
    reg = ToolRegistry()
    hits = []
    lock = threading.Lock()

    def handler(args):
        with lock:
            hits.append(args["i"])
        return {"ok": args["i"]}

    reg.register_tool("T", "t", {"type": "object", "properties": {
        "i": {"type": "integer"}}, "required": ["i"]}, handler)

    errs = []

    def worker(base):
        for i in range(25):
            out = reg.execute_tool("T", {"i": base + i})
            if out.get("ok") != base + i:
                errs.append(out)

    threads = [threading.Thread(target=worker, args=(k * 100,)) for k in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    assert len(hits) == 200


def test_memdir_parallel_creates(memdir_base):
    mu.ensure_folder("", memdir_base)
    errs = []

    def worker(k):
        try:
            for i in range(20):
                mu.create_memory("", {"Subject": f"w{k}-{i}"}, "b",
                                 base=memdir_base)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(k,)) for k in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    mems = mu.list_memories("", "new", base=memdir_base)
    assert len(mems) == 160                    # unique filenames, no clobbers
    assert len({m["filename"] for m in mems}) == 160


def test_chain_parallel_appends(tmp_path):
    from fei_amd.memorychain.chain import MemoryChain
    from fei_amd.memorychain.wallet import FeiCoinWallet

    c = MemoryChain(node_id="n", path=str(tmp_path / "c.json"), difficulty=1,
                    wallet=FeiCoinWallet(path=str(tmp_path / "w.json")))
    errs = []

    def worker(k):
        try:
            for i in range(10):
                c.add_memory(f"m-{k}-{i}", {"Subject": f"{k}-{i}"})
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(k,)) for k in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    assert len(c.blocks) == 41
    assert c.validate_chain()
