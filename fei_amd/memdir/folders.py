"""memdir folder management: CRUD, stats, bulk operations.

Parity: the reference MemdirFolderManager (memdir_tools/folders.py:45-587):
create/rename/delete (with trash evacuation), move/copy between folders,
folder stats (counts, flag/tag histograms, newest/oldest), and bulk tag
rewrite inside memory files.
"""

from __future__ import annotations

import os
import shutil
from collections import Counter
from typing import Any, Dict, List, Optional

from fei_amd.memdir import utils as mu


class MemdirFolderManager:
    def __init__(self, base: Optional[str] = None):
        self.base = mu.get_memdir_base(base)
        mu.ensure_folder("", self.base)

    def _folder_path(self, folder: str) -> str:
        # validated join: rejects names escaping the memdir base
        return mu.folder_path(folder, self.base)

    # -- CRUD ---------------------------------------------------------------

    def create_folder(self, folder: str) -> bool:
        if not folder or folder in ("cur", "new", "tmp"):
            return False
        mu.ensure_folder(folder, self.base)
        return True

    def folder_exists(self, folder: str) -> bool:
        p = self._folder_path(folder)
        return all(os.path.isdir(os.path.join(p, s)) for s in mu.STATUS_DIRS)

    def rename_folder(self, old: str, new: str) -> bool:
        if not self.folder_exists(old) or self.folder_exists(new):
            return False
        dst = self._folder_path(new)
        os.makedirs(os.path.dirname(dst) or self.base, exist_ok=True)
        os.rename(self._folder_path(old), dst)
        return True

    def delete_folder(self, folder: str, force: bool = False) -> bool:
        """Delete a folder; unless force, evacuate its memories to .Trash
        first (reference: folders.py:124-173)."""
        if not self.folder_exists(folder) or not folder:
            return False
        if not force:
            mu.ensure_folder(".Trash", self.base)
            for status in mu.STATUS_DIRS:
                for mem in mu.list_memories(folder, status, base=self.base):
                    mu.move_memory(mem["filename"], folder, ".Trash",
                                   src_status=status, dst_status="cur", base=self.base)
        shutil.rmtree(self._folder_path(folder))
        return True

    # -- move / copy ---------------------------------------------------------

    def move_memories(self, src: str, dst: str, statuses: Optional[List[str]] = None) -> int:
        n = 0
        mu.ensure_folder(dst, self.base)
        for status in statuses or ["cur", "new"]:
            for mem in mu.list_memories(src, status, base=self.base):
                if mu.move_memory(mem["filename"], src, dst, src_status=status,
                                  dst_status=status, base=self.base):
                    n += 1
        return n

    def copy_memory(self, filename: str, src: str, dst: str,
                    src_status: str = "cur", dst_status: str = "cur") -> bool:
        src_path = os.path.join(self._folder_path(src), src_status, filename)
        if not os.path.exists(src_path):
            return False
        mu.ensure_folder(dst, self.base)
        # a copy gets a fresh unique filename (new identity)
        with open(src_path, "r", encoding="utf-8", errors="replace") as f:
            content = f.read()
        headers, body = mu.parse_memory_content(content)
        meta = mu.parse_memory_filename(filename) or {}
        mu.create_memory(dst, headers, body, flags="".join(meta.get("flags", [])),
                         base=self.base, status=dst_status)
        return True

    # -- stats ---------------------------------------------------------------

    def get_folder_stats(self, folder: str = "") -> Dict[str, Any]:
        """Counts per status, flag/tag histograms, newest/oldest timestamps
        (reference: folders.py:216-318)."""
        counts: Dict[str, int] = {}
        flags: Counter = Counter()
        tags: Counter = Counter()
        newest, oldest = 0, 0
        total = 0
        for status in mu.STATUS_DIRS:
            mems = mu.list_memories(folder, status, include_content=True, base=self.base)
            counts[status] = len(mems)
            for m in mems:
                total += 1
                meta = m.get("metadata") or {}
                ts = meta.get("timestamp", 0)
                newest = max(newest, ts)
                oldest = ts if oldest == 0 else min(oldest, ts)
                for fl in meta.get("flags", []):
                    flags[fl] += 1
                tag_header = m.get("headers", {}).get("Tags", "")
                for t in tag_header.replace(",", " ").split():
                    tags[t.strip().lower()] += 1
        return {
            "folder": folder or "(root)",
            "total": total,
            "counts": counts,
            "flags": dict(flags),
            "tags": dict(tags.most_common(20)),
            "newest": newest or None,
            "oldest": oldest or None,
        }

    def list_folders(self) -> List[Dict[str, Any]]:
        out = []
        for folder in mu.list_folders(self.base):
            p = self._folder_path(folder)
            n = sum(len(os.listdir(os.path.join(p, s)))
                    for s in ("cur", "new") if os.path.isdir(os.path.join(p, s)))
            out.append({"folder": folder or "(root)", "count": n})
        return out

    # -- bulk tagging (reference: folders.py:483-587) -------------------------

    def bulk_tag_folder(self, folder: str, tags: List[str], mode: str = "add") -> int:
        """Rewrite the Tags header of every memory in a folder.
        mode: add | remove | replace."""
        n = 0
        for status in ("cur", "new"):
            folder_path = self._folder_path(folder)
            for mem in mu.list_memories(folder, status, include_content=True, base=self.base):
                headers = dict(mem.get("headers", {}))
                current = [t.strip() for t in headers.get("Tags", "").replace(",", " ").split() if t.strip()]
                if mode == "add":
                    new_tags = current + [t for t in tags if t not in current]
                elif mode == "remove":
                    new_tags = [t for t in current if t not in tags]
                else:
                    new_tags = list(tags)
                headers["Tags"] = ",".join(new_tags)
                path = os.path.join(folder_path, status, mem["filename"])
                with open(path, "w", encoding="utf-8") as f:
                    f.write(mu.format_memory_content(headers, mem.get("content", "")))
                n += 1
        return n

    # -- symlinked views (reference: folders.py:382-426) ----------------------

    def link_folder(self, target: str, link_name: str) -> bool:
        """Create a symlinked folder view (e.g. a project alias)."""
        if not self.folder_exists(target) or self.folder_exists(link_name):
            return False
        link_path = self._folder_path(link_name)
        os.makedirs(os.path.dirname(link_path) or self.base, exist_ok=True)
        try:
            os.symlink(self._folder_path(target), link_path,
                       target_is_directory=True)
            return True
        except OSError:
            return False

    def unlink_folder(self, link_name: str) -> bool:
        link_path = self._folder_path(link_name)
        if not os.path.islink(link_path):
            return False
        os.unlink(link_path)
        return True
