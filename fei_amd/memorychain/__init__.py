"""memorychain — distributed memory/task ledger.

Parity contract (SURVEY.md §2.3 / memdir_tools/memorychain.py): the
block/chain JSON schema (SHA-256 block hash, toy proof-of-work, task state
machine), the FeiCoin wallet, the 51 %-quorum proposal consensus, the
longest-chain-with-prefix sync rule, and the HTTP route set under
``/memorychain/*``. One node per GPU-agent in the 8-GPU federation config;
bulk sync can ride RCCL/xGMI between co-located nodes (fei_amd/memorychain/
xgmi_sync.py) while the HTTP control plane stays wire-compatible.
"""

from fei_amd.memorychain.chain import MemoryBlock, MemoryChain, TaskState
from fei_amd.memorychain.wallet import FeiCoinWallet

__all__ = ["MemoryBlock", "MemoryChain", "TaskState", "FeiCoinWallet"]
