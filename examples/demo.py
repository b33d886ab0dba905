#!/usr/bin/env python3
"""Scripted walkthrough of the orchestrator on a FakeEngine pool (no GPU).

Runs a real task end-to-end through the production pipeline — consensus
(2-model pool, fingerprint clustering, merge rules), the action gate chain,
spawn/dismiss, budgets, persistence — with a scripted model that first
writes a file, then spawns a child, then waits.  Prints the agent tree,
action log and cost rollup at the end.

For the same thing over real local models: `python -m quoracle_amd serve
--models llama3-8b#0,llama3-8b#1,llama3-8b#2` (MI355X) and POST to
/api/tasks, or `python -m quoracle_amd serve --fake` for this backend.

Usage: python examples/demo.py
"""

import asyncio
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from quoracle_amd.agent.supervisor import Supervisor
from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.engine.pool import EnginePool
from quoracle_amd.governance.profiles import Profile
from quoracle_amd.tasks.manager import TaskManager
from quoracle_amd.tasks.runtime import RuntimeConfig, TaskRuntime


def scripted(workdir: str):
    """Both pool models agree on each cycle's action (consensus passes
    round 1 unanimously); the plan: write a file -> spawn a child -> wait."""
    def _a(action, params, wait=False):
        return json.dumps({"reasoning": f"demo: {action}", "action": action,
                           "params": params, "wait": wait})
    plan = [
        _a("file_write", {"path": os.path.join(workdir, "notes.md"),
                          "mode": "write",
                          "content": "# demo\nconsensus wrote this."}),
        _a("spawn_child", {"task_description": "investigate the notes",
                           "success_criteria": "notes summarized",
                           "immediate_context": "parent wrote notes.md",
                           "approach_guidance": "read then summarize",
                           "profile": "default", "budget": "2"}),
        _a("wait", {"wait": True}, wait=True),
    ]

    def fn(model_key, messages, request):
        # children (and cycles past the plan) just wait
        step = fn.steps.get(model_key, 0)
        fn.steps[model_key] = step + 1
        if "investigate the notes" in json.dumps(messages):
            return _a("wait", {"wait": True}, wait=True)
        return plan[step] if step < len(plan) else plan[-1]
    fn.steps = {}
    return fn


async def main():
    workdir = tempfile.mkdtemp(prefix="quoracle-demo-")
    engine = FakeEngine(response_fn=scripted(workdir))
    runtime = TaskRuntime(engines=EnginePool(default=engine, embedder=engine),
                          config=RuntimeConfig())
    Supervisor(runtime)
    runtime.profiles.put(Profile(
        name="default", description="demo profile",
        model_pool=["demo-model-a", "demo-model-b"],
        capability_groups=["hierarchy", "local_execution",
                           "file_read", "file_write"]))
    manager = TaskManager(runtime)

    print("== creating task (budget $10) ==")
    created = await manager.create_task(
        "Demonstrate the pipeline", "default", budget_limit=10.0)
    task_id = created["task_id"]
    root_id = created["root_agent_id"]
    print(f"task {task_id} root agent {root_id}")

    # let the root run its plan (file_write -> spawn -> wait)
    for _ in range(200):
        await asyncio.sleep(0.05)
        entry = runtime.registry.lookup(root_id)
        if entry and entry.actor.state.children:
            break

    root = runtime.registry.lookup(root_id).actor
    print("\n== agent tree ==")
    for agent_id in runtime.registry.all_ids():
        e = runtime.registry.lookup(agent_id)
        indent = "  " if e.parent_id else ""
        print(f"{indent}{agent_id}  status={e.actor.state.status}")

    print("\n== artifacts ==")
    notes = os.path.join(workdir, "notes.md")
    print(f"{notes}: {open(notes).read()!r}" if os.path.exists(notes)
          else "file_write pending")

    print("\n== action log (root) ==")
    for row in reversed(runtime.store.logs_for_agent(root_id, limit=10)):
        print(f"  [{row['level']}] {row['event_type']}: "
              f"{row['message'][:90]}")

    print("\n== budget ==")
    print(f"  allocated={root.state.budget_allocated} "
          f"spent={root.state.budget_spent:.4f} "
          f"committed_to_children={root.state.budget_committed}")

    await manager.supervisor.terminate_tree(root_id)
    print("\ndemo complete.")


if __name__ == "__main__":
    asyncio.run(main())
