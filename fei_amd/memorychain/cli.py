"""memorychain CLI.

Command parity with the reference memorychain_cli.py (:44-991):
start / propose / list / view / responsible / connect / status /
network-status / validate / task / tasks / view-task / claim / solve /
vote / difficulty / wallet.
"""

from __future__ import annotations

import argparse
import json
import sys
from typing import List, Optional

from fei_amd.tools.memorychain_connector import MemorychainConnector


def main(argv: Optional[List[str]] = None) -> int:
    p = argparse.ArgumentParser(prog="memorychain",
                                description="memorychain federation CLI")
    p.add_argument("--node", default=None, help="host:port (default env "
                   "MEMORYCHAIN_NODE or localhost:6789)")
    sub = p.add_subparsers(dest="cmd")

    s = sub.add_parser("start", help="run a node")
    s.add_argument("--port", type=int, default=6789)
    s.add_argument("--node-id", default=None)
    s.add_argument("--seed", default=None, help="seed node host:port to join")
    s.add_argument("--chain-path", default=None)
    s.add_argument("--no-persist", action="store_true")

    s = sub.add_parser("propose", help="propose a memory")
    s.add_argument("--subject", required=True)
    s.add_argument("--body", default="")
    s.add_argument("--tags", default="")

    sub.add_parser("list", help="list chain memories")
    s = sub.add_parser("view", help="view a block by memory id")
    s.add_argument("memory_id")
    s = sub.add_parser("responsible", help="memories owned by a node")
    s.add_argument("--for-node", default=None)
    s = sub.add_parser("connect", help="join a network via a seed node")
    s.add_argument("seed")
    sub.add_parser("status", help="node status")
    sub.add_parser("network-status", help="all peers' status")
    sub.add_parser("validate", help="validate the chain")

    s = sub.add_parser("task", help="propose a task")
    s.add_argument("--subject", required=True)
    s.add_argument("--reward", type=float, default=1.0)
    s.add_argument("--difficulty", type=int, default=1)
    s = sub.add_parser("tasks", help="list tasks")
    s.add_argument("--state", default=None)
    s = sub.add_parser("view-task", help="view one task")
    s.add_argument("task_id")
    s = sub.add_parser("claim", help="claim a task")
    s.add_argument("task_id")
    s = sub.add_parser("solve", help="submit a solution")
    s.add_argument("task_id")
    s.add_argument("solution")
    s = sub.add_parser("vote", help="vote on a solution")
    s.add_argument("task_id")
    s.add_argument("--solution-index", type=int, default=0)
    s.add_argument("--reject", action="store_true")
    s = sub.add_parser("difficulty", help="vote on task difficulty")
    s.add_argument("task_id")
    s.add_argument("difficulty", type=int)
    s = sub.add_parser("wallet", help="wallet balance")
    s.add_argument("--for-node", default=None)

    args = p.parse_args(argv)

    if args.cmd == "start":
        from fei_amd.memorychain.node import MemorychainNode
        node_id = args.node_id or f"node-{args.port}"
        node = MemorychainNode(node_id=node_id, port=args.port,
                               chain_path=args.chain_path,
                               persist=not args.no_persist)
        if args.seed:
            ok = node.connect_to_network(args.seed)
            print(f"joined via {args.seed}: {ok}")
        print(f"memorychain node {node_id} on :{args.port}")
        node.run()
        return 0

    conn = MemorychainConnector(node=args.node)

    def dump(obj) -> int:
        print(json.dumps(obj, indent=2, default=str))
        return 0 if not (isinstance(obj, dict) and obj.get("error")) else 1

    if args.cmd == "propose":
        headers = {"Subject": args.subject}
        if args.tags:
            headers["Tags"] = args.tags
        return dump(conn.add_memory(headers, args.body))
    if args.cmd == "list":
        for b in conn.get_chain()[1:]:
            subj = b.get("memory_data", {}).get("Subject", "")
            mark = "T" if b.get("is_task") else " "
            print(f"{b['index']:4d} {mark} {b['memory_id'][:24]:24s} {subj}")
        return 0
    if args.cmd == "view":
        for b in conn.get_chain():
            if b["memory_id"] == args.memory_id or \
                    b["memory_id"].startswith(args.memory_id):
                return dump(b)
        print("not found", file=sys.stderr)
        return 1
    if args.cmd == "responsible":
        url_args = {"node": args.for_node} if args.for_node else {}
        return dump(conn._get("responsible_memories", **url_args))
    if args.cmd == "connect":
        return dump(conn._post("register", {"address": args.seed}))
    if args.cmd == "status":
        return dump(conn._get("node_status"))
    if args.cmd == "network-status":
        return dump(conn.get_network_status())
    if args.cmd == "validate":
        return dump(conn.validate_chain())
    if args.cmd == "task":
        return dump(conn.propose_task({"Subject": args.subject},
                                      reward=args.reward,
                                      difficulty=args.difficulty))
    if args.cmd == "tasks":
        for t in conn.list_tasks(args.state):
            print(f"{t['memory_id'][:24]:24s} {t['task_state']:18s} "
                  f"reward={t.get('reward', 0)} "
                  f"{t.get('memory_data', {}).get('Subject', '')}")
        return 0
    if args.cmd == "view-task":
        return dump(conn._get(f"tasks/{args.task_id}"))
    if args.cmd == "claim":
        return dump(conn.claim_task(args.task_id))
    if args.cmd == "solve":
        return dump(conn.submit_solution(args.task_id, args.solution))
    if args.cmd == "vote":
        return dump(conn._post("vote_solution", {
            "task_id": args.task_id, "solution_index": args.solution_index,
            "approve": not args.reject}))
    if args.cmd == "difficulty":
        return dump(conn._post("vote_difficulty", {
            "task_id": args.task_id, "difficulty": args.difficulty}))
    if args.cmd == "wallet":
        url_args = {"node": args.for_node} if args.for_node else {}
        return dump(conn._get("wallet/balance", **url_args))
    p.print_help()
    return 1


if __name__ == "__main__":
    raise SystemExit(main())
