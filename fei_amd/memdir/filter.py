"""memdir filter engine: email-style rules over new memories.

Parity: the reference MemoryFilter/FilterManager (memdir_tools/filter.py:20-328):
conditions are regexes over a header/content/flags, actions move/flag/copy,
the default processing scope is the ``new`` status, and a set of default
filters is provided.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from fei_amd.memdir import utils as mu
from fei_amd.memdir.folders import MemdirFolderManager


@dataclass
class MemoryFilter:
    name: str
    conditions: List[Dict[str, str]]          # {"field": ..., "pattern": ...}
    actions: List[Dict[str, str]]             # {"action": "move"|"flag"|"copy", ...}
    match_all: bool = True

    def matches(self, mem: Dict[str, Any]) -> bool:
        results = []
        for cond in self.conditions:
            fieldname = cond.get("field", "content")
            pattern = cond.get("pattern", "")
            if fieldname.lower() == "content":
                value = mem.get("content", "")
            elif fieldname.lower() == "flags":
                value = "".join((mem.get("metadata") or {}).get("flags", []))
            else:
                value = mem.get("headers", {}).get(fieldname, "")
            try:
                results.append(re.search(pattern, str(value), re.IGNORECASE) is not None)
            except re.error:
                results.append(False)
        if not results:
            return False
        return all(results) if self.match_all else any(results)

    def apply_actions(self, mem: Dict[str, Any], base: Optional[str] = None) -> List[str]:
        applied = []
        folder = mem.get("folder", "")
        status = mem.get("status", "new")
        filename = mem["filename"]
        for act in self.actions:
            kind = act.get("action")
            if kind == "move":
                target = act.get("folder", "")
                if mu.move_memory(filename, folder, target, src_status=status,
                                  dst_status="cur", base=base):
                    applied.append(f"move:{target}")
                    folder, status = target, "cur"
            elif kind == "flag":
                flags = act.get("flags", "")
                existing = "".join((mem.get("metadata") or {}).get("flags", []))
                new_name = mu.update_memory_flags(filename, folder, status,
                                                  existing + flags, base=base)
                if new_name:
                    applied.append(f"flag:{flags}")
                    filename = new_name
                    mem["filename"] = new_name
            elif kind == "copy":
                target = act.get("folder", "")
                MemdirFolderManager(base).copy_memory(filename, folder, target,
                                                      src_status=status)
                applied.append(f"copy:{target}")
        return applied


def create_default_filters() -> List[MemoryFilter]:
    """Six default rules (reference: filter.py:263-309)."""
    return [
        MemoryFilter("todo-to-later",
                     [{"field": "Subject", "pattern": r"\btodo\b|\blater\b"}],
                     [{"action": "move", "folder": ".ToDoLater"}]),
        MemoryFilter("priority-flag",
                     [{"field": "Priority", "pattern": r"high|urgent"}],
                     [{"action": "flag", "flags": "FP"}]),
        MemoryFilter("project-sort",
                     [{"field": "Tags", "pattern": r"\bproject\b"}],
                     [{"action": "move", "folder": ".Projects"}]),
        MemoryFilter("archive-done",
                     [{"field": "Status", "pattern": r"\bdone\b|\bcompleted\b"}],
                     [{"action": "move", "folder": ".Archive"}]),
        MemoryFilter("flag-important",
                     [{"field": "content", "pattern": r"\bimportant\b"}],
                     [{"action": "flag", "flags": "F"}]),
        MemoryFilter("trash-junk",
                     [{"field": "Subject", "pattern": r"^junk:"}],
                     [{"action": "move", "folder": ".Trash"}]),
    ]


class FilterManager:
    def __init__(self, base: Optional[str] = None, filters: Optional[List[MemoryFilter]] = None):
        self.base = base
        self.filters = filters if filters is not None else create_default_filters()

    def add_filter(self, f: MemoryFilter) -> None:
        self.filters.append(f)

    def process_memories(self, folder: str = "", status: str = "new",
                         move_unmatched_to_cur: bool = True) -> Dict[str, Any]:
        """Run all filters over one folder+status (default scope: new —
        reference: filter.py:175-261). Unmatched new memories graduate to cur."""
        report: Dict[str, Any] = {"processed": 0, "actions": []}
        for mem in mu.list_memories(folder, status, include_content=True, base=self.base):
            report["processed"] += 1
            matched = False
            for f in self.filters:
                if f.matches(mem):
                    actions = f.apply_actions(mem, base=self.base)
                    if actions:
                        matched = True
                        report["actions"].append(
                            {"filter": f.name, "memory": mem["filename"], "applied": actions})
                    # memory may have moved; stop filter chain on move
                    if any(a.startswith("move:") for a in actions):
                        break
            if not matched and status == "new" and move_unmatched_to_cur:
                mu.move_memory(mem["filename"], folder, folder,
                               src_status="new", dst_status="cur", base=self.base)
        return report


def run_filters(base: Optional[str] = None) -> Dict[str, Any]:
    """Process every folder's 'new' memories (reference: filter.py:311-328)."""
    mgr = FilterManager(base)
    total: Dict[str, Any] = {"processed": 0, "actions": []}
    for folder in mu.list_folders(base):
        r = mgr.process_memories(folder)
        total["processed"] += r["processed"]
        total["actions"].extend(r["actions"])
    return total
