"""memdir — Maildir-style on-disk memory system.

On-disk format is byte-compatible with the reference
(memdir_tools/utils.py:59-132): filenames
``{unix_ts}.{uuid4hex[:8]}.{hostname}:2,{FLAGS}``, per-folder ``cur/new/tmp``
status dirs, ``Header: value`` lines + ``---`` + body content, flags S/R/F/P.
"""

from fei_amd.memdir.utils import (
    FLAGS,
    SPECIAL_FOLDERS,
    STATUS_DIRS,
    create_memory,
    get_memdir_base,
    list_memories,
    move_memory,
    parse_memory_content,
    parse_memory_filename,
    update_memory_flags,
)

__all__ = [
    "FLAGS", "SPECIAL_FOLDERS", "STATUS_DIRS", "create_memory",
    "get_memdir_base", "list_memories", "move_memory",
    "parse_memory_content", "parse_memory_filename", "update_memory_flags",
    "EmbeddingIndex", "FtsIndex",
]


def __getattr__(name):
    # lazy: EmbeddingIndex pulls torch (~1.5 s); FtsIndex is cheap but
    # kept symmetrical
    if name == "EmbeddingIndex":
        from fei_amd.memdir.embed_index import EmbeddingIndex
        return EmbeddingIndex
    if name == "FtsIndex":
        from fei_amd.memdir.fts_index import FtsIndex
        return FtsIndex
    raise AttributeError(name)
