"""memdir lifecycle maintenance.

Parity: the reference MemoryArchiver (memdir_tools/archiver.py:45-639):
age-based archiving into ``.Archive/<year>`` subfolders, cleanup rules,
trash expiry (30 d), retention caps scored by age or importance, regex
status rewrite (auto completed/dormant), and a combined run_maintenance.
"""

from __future__ import annotations

import os
import re
import time
from typing import Any, Dict, List, Optional

from fei_amd.memdir import utils as mu

DEFAULT_ARCHIVE_AGE_DAYS = 90
DEFAULT_TRASH_AGE_DAYS = 30


class MemoryArchiver:
    def __init__(self, base: Optional[str] = None):
        self.base = base

    # -- helpers -------------------------------------------------------------

    @staticmethod
    def _age_days(mem: Dict[str, Any]) -> float:
        ts = (mem.get("metadata") or {}).get("timestamp", 0)
        return (time.time() - ts) / 86400.0

    def _year_subfolder(self, mem: Dict[str, Any]) -> str:
        ts = (mem.get("metadata") or {}).get("timestamp", 0)
        year = time.gmtime(ts).tm_year if ts else time.gmtime().tm_year
        return f".Archive/{year}"

    def importance_score(self, mem: Dict[str, Any]) -> float:
        """Flagged/Priority memories and tagged memories score higher
        (reference: archiver.py:433-515)."""
        score = 0.0
        flags = (mem.get("metadata") or {}).get("flags", [])
        if "F" in flags:
            score += 2.0
        if "P" in flags:
            score += 2.0
        if "S" in flags:
            score += 0.5
        headers = mem.get("headers", {})
        if headers.get("Tags"):
            score += 1.0
        prio = headers.get("Priority", "").lower()
        if prio in ("high", "urgent"):
            score += 2.0
        score -= self._age_days(mem) / 365.0
        return score

    # -- operations ----------------------------------------------------------

    def archive_old_memories(self, age_days: float = DEFAULT_ARCHIVE_AGE_DAYS) -> int:
        """Move memories older than ``age_days`` from non-special folders into
        .Archive/<year> (reference: archiver.py:205-304)."""
        n = 0
        for folder in mu.list_folders(self.base):
            if folder.startswith(".Archive") or folder.startswith(".Trash"):
                continue
            for status in ("cur", "new"):
                for mem in mu.list_memories(folder, status, base=self.base):
                    if self._age_days(mem) > age_days:
                        dst = self._year_subfolder(mem)
                        if mu.move_memory(mem["filename"], folder, dst,
                                          src_status=status, dst_status="cur",
                                          base=self.base):
                            n += 1
        return n

    def cleanup_memories(self, rules: Optional[List[Dict[str, Any]]] = None) -> int:
        """Apply cleanup rules: each rule = {folder, max_age_days, action}
        where action is 'trash' or 'delete' (reference: archiver.py:306-381)."""
        rules = rules or [{"folder": ".ToDoLater", "max_age_days": 180, "action": "trash"}]
        n = 0
        for rule in rules:
            folder = rule.get("folder", "")
            max_age = float(rule.get("max_age_days", 365))
            action = rule.get("action", "trash")
            for status in ("cur", "new"):
                for mem in mu.list_memories(folder, status, base=self.base):
                    if self._age_days(mem) <= max_age:
                        continue
                    if action == "delete":
                        root = mu.get_memdir_base(self.base)
                        path = os.path.join(root, folder, status, mem["filename"])
                        try:
                            os.unlink(path)
                            n += 1
                        except OSError:
                            pass
                    else:
                        if mu.move_memory(mem["filename"], folder, ".Trash",
                                          src_status=status, dst_status="cur",
                                          base=self.base):
                            n += 1
        return n

    def empty_trash(self, age_days: float = DEFAULT_TRASH_AGE_DAYS) -> int:
        """Delete trash older than 30 d (reference: archiver.py:383-431)."""
        n = 0
        root = mu.get_memdir_base(self.base)
        for status in mu.STATUS_DIRS:
            for mem in mu.list_memories(".Trash", status, base=self.base):
                if self._age_days(mem) > age_days:
                    try:
                        os.unlink(os.path.join(root, ".Trash", status, mem["filename"]))
                        n += 1
                    except OSError:
                        pass
        return n

    def apply_retention_policies(
        self, max_per_folder: int = 10000, score: str = "importance"
    ) -> int:
        """Cap folder sizes, evicting the lowest-scored memories to .Trash
        (reference: archiver.py:433-515)."""
        n = 0
        for folder in mu.list_folders(self.base):
            if folder.startswith(".Trash"):
                continue
            mems: List[Dict[str, Any]] = []
            for status in ("cur", "new"):
                mems.extend(mu.list_memories(folder, status, include_content=True,
                                             base=self.base))
            if len(mems) <= max_per_folder:
                continue
            if score == "age":
                mems.sort(key=self._age_days)           # newest first stays
            else:
                mems.sort(key=self.importance_score, reverse=True)
            for mem in mems[max_per_folder:]:
                if mu.move_memory(mem["filename"], folder, ".Trash",
                                  src_status=mem["status"], dst_status="cur",
                                  base=self.base):
                    n += 1
        return n

    def update_memory_statuses(self) -> int:
        """Rewrite Status headers: done-keywords -> completed; stale
        in-progress -> dormant (reference: archiver.py:517-619)."""
        n = 0
        root = mu.get_memdir_base(self.base)
        for folder in mu.list_folders(self.base):
            for status in ("cur", "new"):
                for mem in mu.list_memories(folder, status, include_content=True,
                                            base=self.base):
                    headers = dict(mem.get("headers", {}))
                    current = headers.get("Status", "")
                    new_status = None
                    body = mem.get("content", "")
                    if re.search(r"\b(done|finished|completed)\b", body, re.I) and \
                            current.lower() not in ("completed",):
                        new_status = "completed"
                    elif current.lower() == "in-progress" and self._age_days(mem) > 30:
                        new_status = "dormant"
                    if new_status:
                        headers["Status"] = new_status
                        path = os.path.join(root, folder, status, mem["filename"]) \
                            if folder else os.path.join(root, status, mem["filename"])
                        with open(path, "w", encoding="utf-8") as f:
                            f.write(mu.format_memory_content(headers, body))
                        n += 1
        return n

    def run_maintenance(self) -> Dict[str, int]:
        """Combined pass (reference: archiver.py:621-639)."""
        return {
            "archived": self.archive_old_memories(),
            "cleaned": self.cleanup_memories(),
            "trash_emptied": self.empty_trash(),
            "evicted": self.apply_retention_policies(),
            "statuses_updated": self.update_memory_statuses(),
        }
