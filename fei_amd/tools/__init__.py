from fei_amd.tools.registry import Tool, ToolRegistry
from fei_amd.tools.code import create_code_tools

__all__ = ["Tool", "ToolRegistry", "create_code_tools"]
