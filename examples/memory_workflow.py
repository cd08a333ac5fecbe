"""memdir workflow: create, search (query language), semantic search.
Run: MEMDIR_BASE=/tmp/demo_memdir python examples/memory_workflow.py"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.tools.memory_tools import MemoryTools

tools = MemoryTools()
tools.create({"subject": "MI355X attention notes", "tags": "gpu,kernels",
              "body": "split-K decode attention with online softmax"})
tools.create({"subject": "Weekend plans", "tags": "life"})
print("lexical :", [m["headers"]["Subject"]
                    for m in tools.search({"query": "#gpu", "with_content": False})["results"]])
print("indexed :", tools.index_build({}))
print("semantic:", [(m["headers"]["Subject"], m["score"])
                    for m in tools.semantic_search({"query": "GPU attention kernels"})["results"]])
