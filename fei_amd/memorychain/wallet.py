"""FeiCoin wallet: balances + transaction log persisted as JSON.

Parity: reference FeiCoinWallet (memdir_tools/memorychain.py:330-495):
JSON at ``~/.memdir/feicoin_wallet.json``, credit/transfer with a
transaction log, thread-safe.
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Any, Dict, List, Optional

DEFAULT_WALLET_PATH = os.path.join(os.path.expanduser("~"), ".memdir",
                                   "feicoin_wallet.json")


class FeiCoinWallet:
    def __init__(self, path: Optional[str] = None):
        self.path = path or DEFAULT_WALLET_PATH
        self._lock = threading.RLock()
        self.balances: Dict[str, float] = {}
        self.transactions: List[Dict[str, Any]] = []
        self._load()

    def _load(self) -> None:
        try:
            with open(self.path, "r", encoding="utf-8") as f:
                data = json.load(f)
            self.balances = data.get("balances", {})
            self.transactions = data.get("transactions", [])
        except (OSError, json.JSONDecodeError):
            pass

    def _save(self) -> None:
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        tmp = self.path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            json.dump({"balances": self.balances,
                       "transactions": self.transactions[-1000:]}, f, indent=2)
        os.replace(tmp, self.path)

    def balance(self, node_id: str) -> float:
        with self._lock:
            return self.balances.get(node_id, 0.0)

    def credit(self, node_id: str, amount: float, reason: str = "") -> float:
        with self._lock:
            self.balances[node_id] = self.balances.get(node_id, 0.0) + amount
            self.transactions.append({
                "type": "credit", "to": node_id, "amount": amount,
                "reason": reason, "ts": time.time(),
            })
            self._save()
            return self.balances[node_id]

    def transfer(self, src: str, dst: str, amount: float,
                 reason: str = "") -> bool:
        with self._lock:
            if self.balances.get(src, 0.0) < amount:
                return False
            self.balances[src] -= amount
            self.balances[dst] = self.balances.get(dst, 0.0) + amount
            self.transactions.append({
                "type": "transfer", "from": src, "to": dst, "amount": amount,
                "reason": reason, "ts": time.time(),
            })
            self._save()
            return True

    def to_dict(self) -> Dict[str, Any]:
        with self._lock:
            return {"balances": dict(self.balances),
                    "transactions": list(self.transactions[-50:])}
