"""Search-augmented ask (reference: examples/ask_with_search.py): the
assistant answers with code-search tools available, iterating tool rounds
until a plain answer. Scripted backend for a deterministic offline demo —
swap provider="local" on a GPU box."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd.core.assistant import Assistant
from fei_amd.core.backends import ScriptedBackend
from fei_amd.tools.code import create_code_tools
from fei_amd.tools.registry import ToolRegistry

repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
reg = ToolRegistry()
create_code_tools(reg)
script = [
    {"tool_calls": [{"name": "GrepTool",
                     "input": {"pattern": "def attn_decode", "path": repo,
                               "include": "*.py"}}]},
    {"content": "attn_decode lives in fei_amd/ops/__init__.py"},
]
a = Assistant(backend=ScriptedBackend(script), tool_registry=reg)
print(a.ask("where is decode attention dispatched?"))
print("tool rounds:", a.turn_metrics[-1]["rounds"])
