"""Minimal Assistant usage: stub backend on CPU, local engine on GPU.
Run: python examples/basic_chat.py [--local]"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd import Assistant
from fei_amd.tools.code import create_code_tools
from fei_amd.tools.registry import ToolRegistry

provider = "local" if "--local" in sys.argv else "stub"
registry = ToolRegistry()
create_code_tools(registry)
assistant = Assistant(provider=provider, tool_registry=registry)
print(assistant.ask("List the python files in this repo and summarize one."))
