"""Post-mortem task transcript from the store.

CLI sibling of the web monitor's log/mailbox panels (reference:
LogViewLive / MailboxLive): renders a task's agent tree, every consensus
decision, action log and message from the persisted rows — works on any
quoracle.db, live or not.
"""

from __future__ import annotations

import argparse
import time
from typing import Any, Dict, List, Optional

from ..persistence.store import Store


def _fmt_ts(ts: Optional[float]) -> str:
    if not ts:
        return "-"
    return time.strftime("%H:%M:%S", time.localtime(ts))


def render_task(store: Store, task_id: str, *, logs_per_agent: int = 50) -> str:
    task = store.get_task(task_id)
    if task is None:
        return f"unknown task {task_id}"
    out: List[str] = []
    out.append(f"# Task {task_id} [{task.get('status')}]")
    out.append(f"prompt: {task.get('prompt', '')[:300]}")
    if task.get("budget_limit") is not None:
        out.append(f"budget: {task['budget_limit']}")

    agents = store.agents_for_task(task_id)
    children: Dict[Optional[str], List[Dict[str, Any]]] = {}
    for a in agents:
        children.setdefault(a.get("parent_id"), []).append(a)

    def walk(parent: Optional[str], depth: int) -> None:
        for a in children.get(parent, []):
            agent_id = a["agent_id"]
            pad = "  " * depth
            cost = store.cost_rollup(agent_id)
            out.append(f"{pad}- {agent_id} [{a.get('status')}] "
                       f"cost=${cost['total']:.4f}")
            for log in reversed(store.logs_for_agent(agent_id,
                                                     limit=logs_per_agent)):
                msg = str(log.get("message", ""))[:160]
                out.append(f"{pad}    {_fmt_ts(log.get('inserted_at'))} "
                           f"[{log.get('level')}] {log.get('event_type')}: "
                           f"{msg}")
            walk(agent_id, depth + 1)

    out.append("\n## Agent tree")
    walk(None, 0)

    out.append("\n## Messages")
    for m in store.messages_for_task(task_id):
        out.append(f"{_fmt_ts(m.get('inserted_at'))} "
                   f"{m.get('from_agent')} -> {m.get('to_agent')}: "
                   f"{str(m.get('content', ''))[:200]}")
    return "\n".join(out)


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Render a task transcript")
    p.add_argument("db", help="path to the quoracle store (sqlite)")
    p.add_argument("task_id", nargs="?",
                   help="task to render; omitted = list tasks")
    p.add_argument("--logs", type=int, default=50)
    args = p.parse_args(argv)
    store = Store(args.db)
    if not args.task_id:
        for t in store.list_tasks():
            print(f"{t['task_id']}  [{t['status']}]  "
                  f"{str(t.get('prompt', ''))[:80]}")
        return
    print(render_task(store, args.task_id, logs_per_agent=args.logs))


if __name__ == "__main__":
    main()
