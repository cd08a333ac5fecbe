"""Memorychain tests, including the in-process multi-node consensus harness
the reference lacks (SURVEY.md §4 implication)."""

import pytest

from fei_amd.memorychain.chain import MemoryBlock, MemoryChain, TaskState
from fei_amd.memorychain.wallet import FeiCoinWallet


def make_chain(tmp_path, name="n0", **kw):
    return MemoryChain(node_id=name, path=str(tmp_path / f"{name}.json"),
                       difficulty=1,
                       wallet=FeiCoinWallet(path=str(tmp_path / f"{name}_w.json")),
                       **kw)


def test_block_hash_and_pow():
    b = MemoryBlock(index=1, timestamp=1.0, memory_id="m1",
                    memory_data={"Subject": "s"}, proposer_node="n",
                    prev_hash="0")
    h = b.compute_hash()
    b.mine_block(2)
    assert b.hash.startswith("00")
    assert b.hash == b.compute_hash()
    d = b.to_dict()
    b2 = MemoryBlock.from_dict(d)
    assert b2.hash == b.hash and b2.nonce == b.nonce


def test_add_and_validate(tmp_path):
    c = make_chain(tmp_path)
    c.add_memory("m1", {"Subject": "one"})
    c.add_memory("m2", {"Subject": "two"})
    assert len(c.blocks) == 3
    assert c.validate_chain()
    # tamper detection
    c.blocks[1].memory_data["Subject"] = "evil"
    assert not c.validate_chain()


def test_persistence_roundtrip(tmp_path):
    c = make_chain(tmp_path)
    c.add_memory("m1", {"Subject": "persisted"})
    c2 = make_chain(tmp_path)
    assert len(c2.blocks) == 2
    assert c2.blocks[1].memory_data["Subject"] == "persisted"
    assert c2.validate_chain()


def test_single_node_fast_path(tmp_path):
    c = make_chain(tmp_path)
    out = c.propose_memory("m1", {"Subject": "solo"})
    assert out["accepted"] and out["votes"] == 1
    # dedupe rule rejects the same id again
    out = c.propose_memory("m1", {"Subject": "solo"})
    assert not out["accepted"]


class Harness:
    """In-process federation: N chains wired with direct-call transports."""

    def __init__(self, tmp_path, n):
        self.chains = []
        for i in range(n):
            c = make_chain(tmp_path, f"node{i}")
            self.chains.append(c)
        by_addr = {f"addr{i}": c for i, c in enumerate(self.chains)}
        for i, c in enumerate(self.chains):
            c.vote_transport = lambda peer, prop, _me=i: by_addr[peer].vote_on_proposal(prop)
            c.update_transport = lambda peer, blocks, _me=i: by_addr[peer].receive_chain_update(blocks)
            for j in range(len(self.chains)):
                if j != i:
                    c.register_node(f"addr{j}")


def test_federation_quorum_accepts(tmp_path):
    h = Harness(tmp_path, 4)
    out = h.chains[0].propose_memory("m1", {"Subject": "hello net"})
    assert out["accepted"]
    assert out["votes"] == 4 and out["total"] == 4
    # chain update propagated to every peer
    for c in h.chains:
        assert len(c.blocks) == 2
        assert c.blocks[1].memory_id == "m1"


def test_federation_rejects_duplicate(tmp_path):
    h = Harness(tmp_path, 4)
    assert h.chains[0].propose_memory("m1", {"Subject": "x"})["accepted"]
    out = h.chains[1].propose_memory("m1", {"Subject": "x"})
    assert not out["accepted"]          # peers already hold m1 -> no quorum


def test_federation_divergence_rejected(tmp_path):
    """The reference's prefix rule rejects divergent chains
    (memorychain.py:1073-1078) — preserved behavior."""
    h = Harness(tmp_path, 2)
    a, b = h.chains
    # both append locally without consensus -> divergence
    a.add_memory("ma", {"Subject": "a"})
    b.add_memory("mb", {"Subject": "b"})
    assert not b.receive_chain_update(a.serialize())
    a.add_memory("ma2", {"Subject": "a2"})
    assert not b.receive_chain_update(a.serialize())   # longer but divergent


def test_responsible_node_deterministic(tmp_path):
    h = Harness(tmp_path, 3)
    owner0 = h.chains[0].responsible_node("some-memory")
    owner1 = h.chains[1].responsible_node("some-memory")
    assert owner0 == owner1


def test_task_lifecycle_with_reward(tmp_path):
    h = Harness(tmp_path, 3)
    c = h.chains[0]
    out = c.propose_task("t1", {"Subject": "fix bug"}, reward=5.0)
    assert out["accepted"]
    assert c.claim_task("t1", "node1")
    t = c.tasks()[0]
    assert t["task_state"] == TaskState.IN_PROGRESS
    assert "node1" in t["working_nodes"]
    assert c.submit_solution("t1", "patched it", "node1")
    assert c.tasks()[0]["task_state"] == TaskState.SOLUTION_PROPOSED
    r1 = c.vote_on_solution("t1", 0, True, voter="node0")
    assert not r1["completed"]           # 1/3 < 51%
    r2 = c.vote_on_solution("t1", 0, True, voter="node2")
    assert r2["completed"]               # 2/3 >= 51%
    assert c.tasks()[0]["task_state"] == TaskState.COMPLETED
    assert c.wallet.balance("node1") == 5.0
    assert c.validate_chain()


def test_task_difficulty_vote_median(tmp_path):
    c = make_chain(tmp_path)
    c.propose_task("t1", {"Subject": "hard"}, difficulty=1)
    c.vote_on_task_difficulty("t1", 3, voter="a")
    c.vote_on_task_difficulty("t1", 9, voter="b")
    out = c.vote_on_task_difficulty("t1", 5, voter="c")
    assert out["difficulty"] == 5
    assert c.validate_chain()


def test_wallet_transfer(tmp_path):
    w = FeiCoinWallet(path=str(tmp_path / "w.json"))
    w.credit("a", 10.0)
    assert w.transfer("a", "b", 4.0)
    assert w.balance("a") == 6.0 and w.balance("b") == 4.0
    assert not w.transfer("a", "b", 100.0)
    w2 = FeiCoinWallet(path=str(tmp_path / "w.json"))
    assert w2.balance("b") == 4.0


def test_search_and_stats(tmp_path):
    c = make_chain(tmp_path)
    c.add_memory("m1", {"Subject": "GPU kernels", "content": "mfma tiles"})
    c.add_memory("m2", {"Subject": "shopping"})
    assert len(c.search_memories("mfma")) == 1
    s = c.stats()
    assert s["length"] == 3 and s["valid"]


def test_federation_quorum_fails_when_peers_unreachable(tmp_path):
    """Transport failures count as 'no' votes (reference semantics,
    memorychain.py:988-1001): 1 of 4 votes < 51% -> rejected."""
    h = Harness(tmp_path, 4)
    for c in h.chains:
        c.vote_transport = lambda peer, prop: False      # all peers down
    out = h.chains[0].propose_memory("mX", {"Subject": "unlucky"})
    assert not out["accepted"]
    assert out["votes"] == 1 and out["total"] == 4
    assert len(h.chains[0].blocks) == 1                   # nothing appended


def test_federation_vote_transport_exception_safe(tmp_path):
    """A raising transport must not break consensus accounting."""
    h = Harness(tmp_path, 3)

    def boom(peer, prop):
        raise ConnectionError("down")

    h.chains[0].vote_transport = boom
    out = h.chains[0].propose_memory("mY", {"Subject": "s"})
    assert not out["accepted"]           # raising transport = "no" votes
    assert out["votes"] == 1
    assert h.chains[0].validate_chain()


def test_randomized_consensus_soak(tmp_path):
    """Randomized soak over the federation harness: proposals from random
    nodes with randomly failing vote transports. Invariants after every
    accepted round: (a) quorum really held, (b) all reachable nodes carry
    identical chains after propagation, (c) the chain validates, (d) no
    memory id appears twice."""
    import random
    rng = random.Random(1234)
    h = Harness(tmp_path, 5)
    by_addr = {f"addr{i}": c for i, c in enumerate(h.chains)}

    fail_prob = {"addr%d" % i: 0.0 for i in range(5)}

    def flaky_vote(peer, prop):
        if rng.random() < fail_prob[peer]:
            raise ConnectionError("injected partition")
        return by_addr[peer].vote_on_proposal(prop)

    for c in h.chains:
        c.vote_transport = flaky_vote

    accepted = 0
    for round_no in range(40):
        # randomly degrade 0-2 peers for this round
        for k in fail_prob:
            fail_prob[k] = rng.choice([0.0, 0.0, 0.0, 1.0]) \
                if rng.random() < 0.3 else 0.0
        proposer = h.chains[rng.randrange(5)]
        out = proposer.propose_memory(
            f"mem-{round_no}", {"Subject": f"subject {round_no}"})
        if out.get("accepted"):
            accepted += 1
            assert out["votes"] * 2 > out["total"], "accepted without quorum"
    assert accepted >= 10, "soak produced too few accepted rounds"

    # heal the network and converge via anti-entropy sync
    for k in fail_prob:
        fail_prob[k] = 0.0
    longest = max(h.chains, key=lambda c: len(c.blocks))
    for c in h.chains:
        if c is not longest:
            c.receive_chain_update(longest.serialize())
    lengths = {len(c.blocks) for c in h.chains}
    assert len(lengths) == 1, f"nodes diverged: {lengths}"
    tips = {c.blocks[-1].hash for c in h.chains}
    assert len(tips) == 1
    for c in h.chains:
        assert c.validate_chain()
        ids = [b.memory_id for b in c.blocks[1:]]
        assert len(ids) == len(set(ids)), "duplicate memory committed"
