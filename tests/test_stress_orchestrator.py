"""Orchestrator stress: several concurrent tasks with spawn trees, messages
in flight, then pause -> restore -> verify every agent came back with its
history intact. Guards the protocol invariants under real concurrency."""

import asyncio

import pytest

from quoracle_amd.engine.fake import FakeEngine

from helpers import IDLE, POOL2, action_json, make_manager, wait_until

N_TASKS = 4


@pytest.mark.asyncio
async def test_many_tasks_spawn_message_pause_restore():
    engine = FakeEngine(default_response=IDLE, latency_s=0.01)
    manager, runtime = make_manager(engine)

    tasks = []
    for i in range(N_TASKS):
        result = await manager.create_task(f"stress task {i}", "default")
        tasks.append(result)

    # every root spawns one child through the real action path
    for t in tasks:
        root = runtime.registry.lookup(t["root_agent_id"]).actor
        res = await manager.supervisor.spawn_child_action(root, {
            "task_description": "stress child", "success_criteria": "s",
            "immediate_context": "c", "approach_guidance": "a",
            "profile": "default"})
        assert res["status"] == "spawning"

    async def all_children_up():
        return all(runtime.registry.children_of(t["root_agent_id"])
                   for t in tasks)
    assert await wait_until(lambda: asyncio.get_event_loop() and all(
        runtime.registry.children_of(t["root_agent_id"]) for t in tasks),
        timeout=20)

    # concurrent user messages to every task while agents churn
    await asyncio.gather(*[
        manager.send_user_message(t["task_id"], f"ping {j}")
        for j, t in enumerate(tasks)])

    # pause all, verify everything stopped
    for t in tasks:
        await manager.pause_task(t["task_id"])
    for t in tasks:
        assert not runtime.registry.alive(t["root_agent_id"])
        assert runtime.store.get_task(t["task_id"])["status"] == "paused"

    # restore all; every agent (root + child) must come back with history
    for t in tasks:
        out = await manager.restore_task(t["task_id"])
        assert not out["failed"], out
        assert len(out["restored"]) >= 2
    for t in tasks:
        root = runtime.registry.lookup(t["root_agent_id"]).actor
        h = root.state.model_histories[POOL2[0]]
        assert any("stress task" in str(e.get("content", "")) for e in h)
        kids = runtime.registry.children_of(t["root_agent_id"])
        assert kids and runtime.registry.alive(kids[0])

    # everything still responsive after restore
    for j, t in enumerate(tasks):
        assert await manager.send_user_message(t["task_id"], f"post-restore {j}")
    for t in tasks:
        await manager.supervisor.terminate_tree(t["root_agent_id"])
    assert runtime.registry.all_ids() == []
