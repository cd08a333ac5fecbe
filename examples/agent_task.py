"""Autonomous multi-step task with TaskExecutor ([TASK_COMPLETE] loop).
Run: python examples/agent_task.py"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fei_amd import Assistant, TaskExecutor
from fei_amd.tools.code import create_code_tools
from fei_amd.tools.registry import ToolRegistry

registry = ToolRegistry()
create_code_tools(registry)
assistant = Assistant(provider="stub", tool_registry=registry)
result = TaskExecutor(assistant).execute_task(
    "Count the markdown files in this repository.", max_iterations=5)
print(f"complete={result['complete']} iterations={result['iterations']}")
print(result["final_response"])
