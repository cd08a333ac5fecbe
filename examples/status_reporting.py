"""Node status reporting (reference: examples/fei_status_reporting_example.py):
each federation node advertises model/load/current task over HTTP;
update_status mutates it and node_status/network_status report it."""
import os, sys, tempfile
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.memorychain.node import MemorychainNode

with tempfile.TemporaryDirectory() as d:
    node = MemorychainNode("gpu0", port=6789, difficulty=1,
                           chain_path=os.path.join(d, "c.json"),
                           ai_model="llama3-8b-local")
    c = node.app.test_client()
    print("before:", c.get("/memorychain/node_status").get_json())
    c.post("/memorychain/update_status",
           json={"status": "working", "load": 0.42,
                 "current_task": "review PR #7"})
    st = c.get("/memorychain/node_status").get_json()
    print("after :", {k: st[k] for k in
                      ("node_id", "status", "load", "current_task", "ai_model")})
