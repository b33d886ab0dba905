"""Shared test fixtures: isolated runtime + scripted engines.

Mirrors the reference's isolation-by-injection strategy: every
test builds its own runtime (bus, registry, store, engine pool) — no globals.
"""

import json

from quoracle_amd.agent.supervisor import Supervisor
from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.engine.pool import EnginePool
from quoracle_amd.governance.profiles import Profile
from quoracle_amd.tasks.manager import TaskManager
from quoracle_amd.tasks.runtime import RuntimeConfig, TaskRuntime

POOL2 = ["fake-a", "fake-b"]
POOL3 = ["fake-a", "fake-b", "fake-c"]

IDLE = json.dumps({"reasoning": "idle", "action": "wait",
                   "params": {"wait": True}, "wait": True})


def action_json(action, params=None, reasoning="r", wait=False, **extra):
    body = {"reasoning": reasoning, "action": action, "params": params or {},
            "wait": wait}
    body.update(extra)
    return json.dumps(body)


def make_runtime(engine=None, models=POOL2, config=None, store=None):
    engine = engine or FakeEngine(default_response=IDLE)
    pool = EnginePool(default=engine, embedder=engine)
    runtime = TaskRuntime(engines=pool, config=config or RuntimeConfig(),
                          store=store)
    Supervisor(runtime)
    runtime.profiles.put(Profile(
        name="default", description="test profile", model_pool=list(models),
        capability_groups=["hierarchy", "local_execution", "file_read",
                           "file_write", "external_api"]))
    return runtime


def make_manager(engine=None, models=POOL2, config=None, store=None):
    runtime = make_runtime(engine, models, config, store)
    return TaskManager(runtime), runtime


async def wait_until(predicate, timeout=5.0, interval=0.01):
    import asyncio
    import time
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if predicate():
            return True
        await asyncio.sleep(interval)
    return False
