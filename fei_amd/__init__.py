"""fei_amd — an MI355X-native code-assistant framework.

A from-scratch rebuild of the capabilities of the reference agent
(``david-strejc/fei``): an LLM-driven code assistant with a tool-calling
loop, a Maildir-style memory system ("memdir"), a distributed memory/task
ledger ("memorychain"), and two UIs — except that where the reference calls
remote LLM APIs through LiteLLM, this framework runs a *local* inference
engine on AMD Instinct MI355X GPUs: hand-written CDNA4 (gfx950) HIP kernels
for RMSNorm, RoPE, prefill/decode attention, SwiGLU and sampling, hipGraph
captured decode steps, and tensor parallelism over RCCL/xGMI.

Public API parity targets:
  - ``fei_amd.core.Assistant`` with ``chat()`` and ``ask()``
    (reference: fei/core/assistant.py:320,440)
  - memdir on-disk format + query language
    (reference: memdir_tools/utils.py:59-132, memdir_tools/search.py:392-519)
  - memorychain block/chain JSON schema + HTTP routes
    (reference: memdir_tools/memorychain.py:263-327,1263-1685)
"""

__version__ = "0.1.0"

from fei_amd.core.assistant import Assistant  # noqa: F401
from fei_amd.core.task_executor import TaskExecutor  # noqa: F401

__all__ = ["Assistant", "TaskExecutor", "LocalEngine",
           "PagedSessionManager", "__version__"]


def __getattr__(name):
    # lazy: keep `import fei_amd` torch-free (torch costs ~1.5 s and the
    # agent/memdir layers don't need it)
    if name == "LocalEngine":
        from fei_amd.engine.engine import LocalEngine
        return LocalEngine
    if name == "PagedSessionManager":
        from fei_amd.engine.sessions import PagedSessionManager
        return PagedSessionManager
    raise AttributeError(name)
