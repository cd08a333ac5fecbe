from fei_amd.core.assistant import Assistant, ConversationManager, ProviderManager, ToolManager
from fei_amd.core.task_executor import TaskExecutor, TaskContext

__all__ = [
    "Assistant", "ConversationManager", "ProviderManager", "ToolManager",
    "TaskExecutor", "TaskContext",
]
