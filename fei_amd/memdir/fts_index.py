"""Lexical full-text index over the memdir corpus (SQLite FTS5).

The reference's keyword search re-reads every memory file per query
(memdir_tools/search.py — O(corpus) file IO). The semantic index
(embed_index.py) replaces that for similarity queries; THIS index covers
ranked keyword/boolean search: one bm25 query against
``<base>/.index/fts.db`` instead of a tree walk. Incremental add/remove
mirrors the embedding index so both stay live as memories are created and
deleted; the grammar-based scan (search.py) remains the exact-semantics
path for operator queries (#tag, +F, field:value, /regex/).
"""

from __future__ import annotations

import os
import sqlite3
from typing import Any, Dict, List, Optional, Tuple

from fei_amd.memdir import utils as mu
from fei_amd.utils.logging import get_logger

logger = get_logger("memdir.fts_index")

_SCHEMA = """
CREATE VIRTUAL TABLE IF NOT EXISTS mem_fts USING fts5(
    key UNINDEXED, subject, tags, content, tokenize='porter unicode61');
"""


class FtsIndex:
    def __init__(self, base: Optional[str] = None):
        self.base = mu.get_memdir_base(base)
        self.index_dir = os.path.join(self.base, ".index")
        os.makedirs(self.index_dir, exist_ok=True)
        self.db_path = os.path.join(self.index_dir, "fts.db")
        self._db = sqlite3.connect(self.db_path)
        self._db.executescript(_SCHEMA)

    # -- building ------------------------------------------------------------

    def build(self, folders: Optional[List[str]] = None,
              statuses: Optional[List[str]] = None) -> int:
        """(Re)build from the memdir tree; returns the number of rows."""
        self._db.execute("DELETE FROM mem_fts")
        n = 0
        for folder in folders if folders is not None else mu.list_folders(self.base):
            for status in statuses if statuses is not None else ["cur", "new"]:
                for mem in mu.list_memories(folder, status,
                                            include_content=True,
                                            base=self.base):
                    self._insert(folder, status, mem)
                    n += 1
        self._db.commit()
        logger.info("fts indexed %d memories", n)
        return n

    # key separator: 0x1f (unit separator), NOT NUL — SQLite's LIKE and
    # other text functions stop at an embedded NUL, which silently broke
    # remove() on root-folder keys

    def _insert(self, folder: str, status: str, mem: Dict[str, Any]) -> None:
        headers = mem.get("headers", {})
        key = f"{folder}\x1f{status}\x1f{mem['filename']}"
        self._db.execute(
            "INSERT INTO mem_fts (key, subject, tags, content) VALUES (?,?,?,?)",
            (key, headers.get("Subject", ""), headers.get("Tags", ""),
             (mem.get("content") or "")[:20000]))

    def add(self, folder: str, status: str, filename: str, subject: str,
            tags: str, content: str) -> None:
        key = f"{folder}\x1f{status}\x1f{filename}"
        self._db.execute(
            "INSERT INTO mem_fts (key, subject, tags, content) VALUES (?,?,?,?)",
            (key, subject, tags, content[:20000]))
        self._db.commit()

    def remove(self, key_substr: str) -> int:
        # virtual tables don't report rowcount; count explicitly
        before = self.count()
        self._db.execute(
            "DELETE FROM mem_fts WHERE key LIKE ?", (f"%{key_substr}%",))
        self._db.commit()
        return before - self.count()

    def count(self) -> int:
        return self._db.execute("SELECT COUNT(*) FROM mem_fts").fetchone()[0]

    # -- querying ------------------------------------------------------------

    @staticmethod
    def _fts_query(query: str) -> str:
        """Plain keywords -> implicit AND of quoted terms (so punctuation
        and FTS operators in user text cannot break the query)."""
        terms = [t.replace('"', '') for t in query.split() if t.strip('"')]
        return " ".join(f'"{t}"' for t in terms) or '""'

    def search(self, query: str, limit: int = 20) -> List[Tuple[str, float]]:
        """bm25-ranked [(key, score)] — lower score = better (bm25)."""
        try:
            rows = self._db.execute(
                "SELECT key, bm25(mem_fts, 5.0, 3.0, 1.0) AS r FROM mem_fts "
                "WHERE mem_fts MATCH ? ORDER BY r LIMIT ?",
                (self._fts_query(query), limit)).fetchall()
        except sqlite3.OperationalError:
            return []
        return [(k, float(r)) for k, r in rows]

    def search_memories(self, query: str, limit: int = 20,
                        with_content: bool = True) -> List[Dict[str, Any]]:
        out = []
        for key, score in self.search(query, limit):
            folder, status, filename = key.split("\x1f")
            mem = mu.read_memory(folder, status, filename, base=self.base)
            if mem is None:
                continue
            mem["score"] = round(score, 4)
            if not with_content:
                mem.pop("content", None)
            out.append(mem)
        return out

    def close(self) -> None:
        self._db.close()
