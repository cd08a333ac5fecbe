"""memdir CLI: create/list/view/move/search/flag/mkdir
(reference parity: memdir_tools/cli.py:69-408)."""

from __future__ import annotations

import argparse
import sys
from typing import List, Optional

from fei_amd.memdir import utils as mu
from fei_amd.memdir import search as msearch
from fei_amd.memdir.folders import MemdirFolderManager


def main(argv: Optional[List[str]] = None) -> int:
    parser = argparse.ArgumentParser(prog="memdir", description="memdir memory CLI")
    parser.add_argument("--base", default=None)
    sub = parser.add_subparsers(dest="cmd")

    p = sub.add_parser("create", help="create a memory")
    p.add_argument("--subject", required=True)
    p.add_argument("--body", default="")
    p.add_argument("--folder", default="")
    p.add_argument("--tags", default="")
    p.add_argument("--flags", default="")

    p = sub.add_parser("list", help="list memories")
    p.add_argument("--folder", default="")
    p.add_argument("--status", default="cur")

    p = sub.add_parser("view", help="view one memory")
    p.add_argument("memory_id")

    p = sub.add_parser("move", help="move a memory")
    p.add_argument("memory_id")
    p.add_argument("target_folder")

    p = sub.add_parser("search", help="search with the query language")
    p.add_argument("query", nargs="+")
    p.add_argument("--format", default="text", choices=["text", "json", "csv", "compact"])
    p.add_argument("--fts", action="store_true",
                   help="bm25 FTS index instead of the grammar scan "
                        "(build: memdir index)")

    sub.add_parser("index", help="(re)build the lexical FTS index")

    p = sub.add_parser("flag", help="set flags on a memory")
    p.add_argument("memory_id")
    p.add_argument("flags")

    p = sub.add_parser("mkdir", help="create a folder")
    p.add_argument("folder")

    p = sub.add_parser("folders", help="list folders")

    args = parser.parse_args(argv)
    base = args.base

    if args.cmd == "create":
        headers = {"Subject": args.subject}
        if args.tags:
            headers["Tags"] = args.tags
        name = mu.create_memory(args.folder, headers, args.body, args.flags, base=base)
        print(name)
        return 0
    if args.cmd == "list":
        for mem in mu.list_memories(args.folder, args.status, include_content=True, base=base):
            meta = mem["metadata"]
            subj = mem.get("headers", {}).get("Subject", "(no subject)")
            print(f"{meta['unique']}  {''.join(meta['flags']):4s}  {subj}")
        return 0
    if args.cmd == "view":
        loc = mu.find_memory(args.memory_id, base=base)
        if loc is None:
            print("not found", file=sys.stderr)
            return 1
        mem = mu.read_memory(*loc, base=base)
        for k, v in mem["headers"].items():
            print(f"{k}: {v}")
        print("---")
        print(mem["content"])
        return 0
    if args.cmd == "move":
        loc = mu.find_memory(args.memory_id, base=base)
        if loc is None:
            print("not found", file=sys.stderr)
            return 1
        folder, status, filename = loc
        ok = mu.move_memory(filename, folder, args.target_folder,
                            src_status=status, dst_status="cur", base=base)
        print("moved" if ok else "failed")
        return 0 if ok else 1
    if args.cmd == "search":
        if args.fts:
            from fei_amd.memdir.fts_index import FtsIndex
            results = FtsIndex(base=base).search_memories(" ".join(args.query))
        else:
            results = msearch.search(" ".join(args.query), base=base)
        print(msearch.format_results(results, args.format))
        return 0
    if args.cmd == "index":
        from fei_amd.memdir.fts_index import FtsIndex
        n = FtsIndex(base=base).build()
        print(f"indexed {n} memories")
        return 0
    if args.cmd == "flag":
        loc = mu.find_memory(args.memory_id, base=base)
        if loc is None:
            print("not found", file=sys.stderr)
            return 1
        folder, status, filename = loc
        new_name = mu.update_memory_flags(filename, folder, status, args.flags, base=base)
        print(new_name or "failed")
        return 0 if new_name else 1
    if args.cmd == "mkdir":
        ok = MemdirFolderManager(base).create_folder(args.folder)
        print("created" if ok else "failed")
        return 0 if ok else 1
    if args.cmd == "folders":
        for f in MemdirFolderManager(base).list_folders():
            print(f"{f['count']:6d}  {f['folder']}")
        return 0
    parser.print_help()
    return 1


if __name__ == "__main__":
    raise SystemExit(main())
