"""memdir semantic search: a persisted embedding index over the corpus.

Replaces the reference's O(corpus) per-query full file scan
(memdir_tools/search.py:361-367) for semantic retrieval: embeddings are
computed by the MI355X encoder (fei_amd/models/bge.py — GEMMs on MFMA),
persisted next to the Memdir tree, and queried with one GEMV + top-k on
GPU (BASELINE.json configs[3]: 1M-memory corpus, 1 GPU). The lexical query
language stays available for format compatibility; ``search_semantic``
fuses both: candidates by cosine then optional lexical filtering.

Index layout on disk (under ``<base>/.index/``):
  embeddings.npy   float16 [N, C]
  meta.json        {"ids": [...], "dim": C, "version": 1}
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from fei_amd.memdir import utils as mu
from fei_amd.utils.logging import get_logger

logger = get_logger("memdir.embed_index")


class EmbeddingIndex:
    def __init__(self, base: Optional[str] = None, encoder=None,
                 device: Optional[torch.device] = None):
        self.base = mu.get_memdir_base(base)
        self.index_dir = os.path.join(self.base, ".index")
        self._encoder = encoder
        if device is None:
            device = torch.device("cuda:0") if torch.cuda.is_available() \
                else torch.device("cpu")
        self.device = device
        self.embeddings: Optional[torch.Tensor] = None    # [N, C] on device
        self.ids: List[str] = []                          # "folder/status/filename"
        self._loaded = False

    # -- encoder -------------------------------------------------------------

    @property
    def encoder(self):
        if self._encoder is None:
            from fei_amd.models.bge import BgeEncoder
            self._encoder = BgeEncoder(device=self.device)
        return self._encoder

    # -- persistence ---------------------------------------------------------

    def save(self) -> None:
        os.makedirs(self.index_dir, exist_ok=True)
        if self.embeddings is None:
            return
        np.save(os.path.join(self.index_dir, "embeddings.npy"),
                self.embeddings.cpu().to(torch.float16).numpy())
        with open(os.path.join(self.index_dir, "meta.json"), "w") as f:
            json.dump({"ids": self.ids, "dim": self.embeddings.shape[1],
                       "version": 1}, f)

    def load(self) -> bool:
        try:
            emb = np.load(os.path.join(self.index_dir, "embeddings.npy"))
            with open(os.path.join(self.index_dir, "meta.json")) as f:
                meta = json.load(f)
            self.embeddings = torch.from_numpy(emb).to(self.device).float()
            self.ids = meta["ids"]
            self._loaded = True
            return True
        except (OSError, json.JSONDecodeError, ValueError):
            return False

    # -- building ------------------------------------------------------------

    @staticmethod
    def _memory_text(mem: Dict[str, Any]) -> str:
        headers = mem.get("headers", {})
        return (headers.get("Subject", "") + "\n" +
                headers.get("Tags", "") + "\n" +
                mem.get("content", ""))[:2000]

    def build(self, folders: Optional[List[str]] = None,
              statuses: Optional[List[str]] = None,
              batch_size: int = 64) -> int:
        """(Re)build the index over the memdir corpus. Off the query
        critical path by design (SURVEY.md §7 hard-part 4)."""
        t0 = time.time()
        texts: List[str] = []
        ids: List[str] = []
        for folder in folders if folders is not None else mu.list_folders(self.base):
            for status in statuses if statuses is not None else ["cur", "new"]:
                for mem in mu.list_memories(folder, status, include_content=True,
                                            base=self.base):
                    texts.append(self._memory_text(mem))
                    ids.append(f"{folder}\x00{status}\x00{mem['filename']}")
        if not texts:
            self.embeddings = None
            self.ids = []
            return 0
        emb = self.encoder.encode_texts(texts, batch_size=batch_size)
        self.embeddings = emb.float()
        self.ids = ids
        self._loaded = True
        self.save()
        logger.info("indexed %d memories in %.2fs", len(ids), time.time() - t0)
        return len(ids)

    def remove(self, id_substr: str) -> int:
        """Drop rows whose id contains ``id_substr`` (e.g. a filename when
        the memory was moved/deleted). Returns how many were removed."""
        if self.embeddings is None:
            return 0
        keep = [i for i, k in enumerate(self.ids) if id_substr not in k]
        removed = len(self.ids) - len(keep)
        if removed:
            self.embeddings = self.embeddings[keep]
            self.ids = [self.ids[i] for i in keep]
        return removed

    def add_texts(self, texts: List[str], ids: List[str]) -> None:
        """Incremental append (used by tests and live updates)."""
        emb = self.encoder.encode_texts(texts).float()
        if self.embeddings is None:
            self.embeddings = emb
            self.ids = list(ids)
        else:
            self.embeddings = torch.cat([self.embeddings, emb], dim=0)
            self.ids.extend(ids)

    # -- querying ------------------------------------------------------------

    def search(self, query: str, topk: int = 10) -> List[Tuple[str, float]]:
        """Cosine top-k over the index: one GEMV (MFMA via rocBLAS on GPU)
        + torch.topk. Returns [(id, score)]."""
        if self.embeddings is None and not self._loaded:
            self.load()                    # fall back to the on-disk index
        if self.embeddings is None or not len(self.ids):
            return []
        q = self.encoder.encode_texts([query])[0]          # [C], normalised
        scores = self.embeddings @ q                        # [N]
        k = min(topk, scores.shape[0])
        vals, idx = torch.topk(scores, k)
        return [(self.ids[int(i)], float(v)) for v, i in zip(vals, idx)]

    def search_memories(self, query: str, topk: int = 10,
                        with_content: bool = True) -> List[Dict[str, Any]]:
        """Top-k resolved back to memdir records with scores."""
        out = []
        for key, score in self.search(query, topk):
            folder, status, filename = key.split("\x00")
            mem = mu.read_memory(folder, status, filename, base=self.base)
            if mem is None:
                continue
            mem["score"] = round(score, 4)
            if not with_content:
                mem.pop("content", None)
            out.append(mem)
        return out
