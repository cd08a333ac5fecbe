"""8-node memorychain federation in one process: propose, task lifecycle,
reward payout. Run: python examples/federation_demo.py"""
import os, sys, tempfile
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.memorychain.chain import MemoryChain
from fei_amd.memorychain.wallet import FeiCoinWallet

tmp = tempfile.mkdtemp()
chains = [MemoryChain(node_id=f"node{i}", path=f"{tmp}/c{i}.json", difficulty=1,
                      wallet=FeiCoinWallet(path=f"{tmp}/w{i}.json"))
          for i in range(8)]
by_addr = {f"addr{i}": c for i, c in enumerate(chains)}
for i, c in enumerate(chains):
    c.vote_transport = lambda p, prop: by_addr[p].vote_on_proposal(prop)
    c.update_transport = lambda p, blocks: by_addr[p].receive_chain_update(blocks)
    for j in range(8):
        if j != i:
            c.register_node(f"addr{j}")

out = chains[0].propose_memory("shared-1", {"Subject": "hello federation"})
print("consensus:", out)
chains[1].propose_task("fix-bug-7", {"Subject": "fix the flaky test"}, reward=3.0)
chains[1].claim_task("fix-bug-7", "node2")
chains[1].submit_solution("fix-bug-7", "patched in commit abc", "node2")
for voter in ("node0", "node3", "node4", "node5", "node6"):
    r = chains[1].vote_on_solution("fix-bug-7", 0, True, voter=voter)
print("task:", chains[1].tasks()[0]["task_state"],
      "| node2 balance:", chains[1].wallet.balance("node2"))
