"""BASELINE config[5]: memorychain federation — consensus latency.

In-process 8-node federation (the same chain/consensus code the HTTP node
serves; transports wired directly so this measures consensus + mining +
chain-propagation work, not socket overhead). The reference has no
published numbers; this sets them.

Prints one JSON line: propose->quorum->propagate latency p50/p95 over 100
proposals, 8 nodes.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.memorychain.wallet import FeiCoinWallet


def main(n_nodes: int = 8, n_proposals: int = 100):
    import tempfile
    tmp = tempfile.mkdtemp(prefix="fedbench")
    chains = []
    for i in range(n_nodes):
        chains.append(MemoryChain(
            node_id=f"node{i}", path=os.path.join(tmp, f"c{i}.json"),
            difficulty=2,
            wallet=FeiCoinWallet(path=os.path.join(tmp, f"w{i}.json"))))
    by_addr = {f"addr{i}": c for i, c in enumerate(chains)}
    for i, c in enumerate(chains):
        c.vote_transport = lambda peer, prop: by_addr[peer].vote_on_proposal(prop)
        c.update_transport = lambda peer, blocks: by_addr[peer].receive_chain_update(blocks)
        for j in range(n_nodes):
            if j != i:
                c.register_node(f"addr{j}")

    lat = []
    for k in range(n_proposals):
        proposer = chains[k % n_nodes]
        t0 = time.perf_counter()
        out = proposer.propose_memory(f"bench-mem-{k}",
                                      {"Subject": f"benchmark memory {k}"})
        lat.append(time.perf_counter() - t0)
        assert out["accepted"], out
    assert all(len(c.blocks) == n_proposals + 1 for c in chains)
    assert all(c.validate_chain() for c in chains)
    lat.sort()
    print(json.dumps({
        "metric": "memorychain consensus latency (8 nodes, propose->quorum->sync)",
        "unit": "ms",
        "p50": round(lat[len(lat) // 2] * 1000, 2),
        "p95": round(lat[int(len(lat) * 0.95)] * 1000, 2),
        "proposals": n_proposals, "nodes": n_nodes,
        "chain_valid_everywhere": True,
    }))


if __name__ == "__main__":
    main()
