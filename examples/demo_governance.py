#!/usr/bin/env python3
"""Governance walkthrough: grove hard rules, confinement, and secret
scrubbing through the production gate chain (no GPU).

The scripted pool tries, in order: a blocked shell command (grove hard
rule), a file write outside confinement, a secret generation + use whose
value must be scrubbed from results, and finally an allowed confined
write.  Run: python examples/demo_governance.py
"""

import asyncio
import os

# the vault refuses to exist without key material (docs/SECURITY.md);
# a deployment sets this in its environment
os.environ.setdefault("QUORACLE_VAULT_KEY", "demo-governance-key")
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from quoracle_amd.agent.supervisor import Supervisor
from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.engine.pool import EnginePool
from quoracle_amd.governance.groves import load_grove
from quoracle_amd.governance.profiles import Profile
from quoracle_amd.tasks.manager import TaskManager
from quoracle_amd.tasks.runtime import RuntimeConfig, TaskRuntime

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def scripted():
    def _a(action, params):
        return json.dumps({"reasoning": f"demo: {action}", "action": action,
                           "params": params, "wait": False})
    grove_dir = os.path.join(REPO, "groves", "code-sandbox")
    plan = [
        _a("execute_shell", {"command": "curl http://example.com"}),
        _a("file_write", {"path": "/etc/evil.txt", "mode": "write",
                          "content": "outside confinement"}),
        _a("generate_secret", {"name": "api_token", "length": 24}),
        _a("execute_shell", {"command": "echo token={{SECRET:api_token}}",
                             "working_dir": os.path.join(grove_dir,
                                                         "workspace")}),
        _a("file_write", {"path": os.path.join(grove_dir, "workspace",
                                               "demo-note.txt"),
                          "mode": "write", "content": "confined write ok"}),
        json.dumps({"reasoning": "done", "action": "wait",
                    "params": {"wait": True}, "wait": True}),
    ]

    def fn(model_key, messages, request):
        step = fn.steps.get(model_key, 0)
        fn.steps[model_key] = step + 1
        return plan[min(step, len(plan) - 1)]
    fn.steps = {}
    return fn


async def main():
    engine = FakeEngine(response_fn=scripted())
    runtime = TaskRuntime(engines=EnginePool(default=engine, embedder=engine),
                          config=RuntimeConfig())
    Supervisor(runtime)
    runtime.profiles.put(Profile(
        name="default", description="governance demo",
        model_pool=["m-a", "m-b"],
        capability_groups=["local_execution", "file_read", "file_write"]))
    manager = TaskManager(runtime)
    grove = load_grove(os.path.join(REPO, "groves", "code-sandbox"))
    created = await manager.create_task("governance walkthrough", "default",
                                        grove=grove)
    root_id = created["root_agent_id"]

    deadline = asyncio.get_event_loop().time() + 20
    while asyncio.get_event_loop().time() < deadline:
        await asyncio.sleep(0.1)
        logs = runtime.store.logs_for_agent(root_id, limit=50)
        if sum("action_" in r["event_type"] for r in logs) >= 5:
            break

    print("== gate-chain results (oldest first) ==")
    actor = runtime.registry.lookup(root_id).actor
    for entry in reversed(actor.state.model_histories["m-a"]):
        if entry.get("type") != "result":
            continue
        text = str(entry.get("content", ""))
        head = text.splitlines()[0] if text else ""
        body = " ".join(text.splitlines()[1:])[:84]
        print(f"  {head[:40]:42s} {body}")

    note = os.path.join(REPO, "groves", "code-sandbox", "workspace",
                        "demo-note.txt")
    print(f"\nconfined write landed: {os.path.exists(note)}")
    if os.path.exists(note):
        os.remove(note)

    # scrubbing: the secret value must not appear in any persisted result
    secret = runtime.vault.get("api_token")
    blob = json.dumps(runtime.store.logs_for_agent(root_id, limit=50))
    hist = json.dumps(actor.state.model_histories, default=str)
    print(f"secret leaked into logs/history: "
          f"{secret in blob or secret in hist} "
          f"(REDACTED marker present: {'[REDACTED:api_token]' in hist})")

    await manager.supervisor.terminate_tree(root_id)
    print("\ndemo complete.")


if __name__ == "__main__":
    asyncio.run(main())
