"""Protocol-race semantics: messages arriving mid-cycle
queue and batch into the NEXT cycle as a simultaneous block; dismiss-vs-
spawn guard; stale wait-timer generations."""

import asyncio

import pytest

from quoracle_amd.engine.fake import FakeEngine

from helpers import IDLE, POOL2, action_json, make_manager, wait_until


@pytest.mark.asyncio
async def test_messages_during_cycle_batch_into_next():
    engine = FakeEngine(default_response=IDLE, latency_s=0.25)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("busy agent", "default")
    root_id = result["root_agent_id"]
    actor = runtime.registry.lookup(root_id).actor
    # while cycle 1 is in flight (engine latency), deliver two messages
    await asyncio.sleep(0.05)
    await actor.deliver({"type": "user_message", "content": "first mid-cycle"})
    await actor.deliver({"type": "user_message", "content": "second mid-cycle"})
    assert await wait_until(lambda: actor.steps_completed >= 2, timeout=15)

    h = actor.state.model_histories[POOL2[0]]
    joined = [str(e.get("content", "")) for e in h]
    batched = [c for c in joined if "simultaneous_messages" in c]
    assert batched, "mid-cycle messages were not batch-flushed"
    assert "first mid-cycle" in batched[0] and "second mid-cycle" in batched[0]


@pytest.mark.asyncio
async def test_dismiss_vs_spawn_race_guard():
    """A spawn attempted while the parent has a dismissal in flight is
    refused (reference: spawn.ex:73-107 returns :parent_dismissing).  The
    race is created for real: dismiss_child_action marks the child as
    dismissing synchronously and finishes in a background task, so a spawn
    issued before that task runs hits the open window."""
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("race", "default")
    root = runtime.registry.lookup(result["root_agent_id"]).actor
    spawn_params = {"task_description": "t", "success_criteria": "s",
                    "immediate_context": "c", "approach_guidance": "a",
                    "profile": "default"}
    first = await manager.supervisor.spawn_child_action(root, spawn_params)
    child_id = first["child_id"]
    assert child_id in root.state.children

    # dismissal marked synchronously; background teardown has NOT run yet
    res = await manager.supervisor.dismiss_child_action(root, child_id, "r")
    assert res.get("status") == "dismissing"
    assert root.state.dismissing

    blocked = await manager.supervisor.spawn_child_action(root, spawn_params)
    assert blocked.get("error") == "parent_dismissing"
    assert len(root.state.children) <= 1  # no second child appeared

    # once the dismissal completes, spawning works again
    for _ in range(50):
        if not root.state.dismissing:
            break
        await asyncio.sleep(0.01)
    assert not root.state.dismissing
    after = await manager.supervisor.spawn_child_action(root, spawn_params)
    assert "error" not in after
    await manager.supervisor.terminate_tree(root.state.agent_id)


@pytest.mark.asyncio
async def test_stale_wait_timer_does_not_retrigger():
    """A result arriving before the wait timer bumps the generation; the
    old timer firing later must not cause a spurious extra cycle
    (reference: state.ex:87-88 wait generations)."""
    seq = [action_json("wait", {"wait": 1}, wait=1), IDLE]
    state = {"i": 0}

    def responder(model, msgs, req):
        idx = min(state["i"] // 2, len(seq) - 1)
        state["i"] += 1
        return seq[idx]

    engine = FakeEngine(response_fn=responder)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("timer race", "default")
    root_id = result["root_agent_id"]
    actor = runtime.registry.lookup(root_id).actor
    assert await wait_until(lambda: actor.steps_completed >= 1, timeout=10)
    gen_before = actor.state.wait_generation
    # a real message preempts the pending timer
    await manager.send_user_message(result["task_id"], "preempt the timer")
    assert await wait_until(lambda: actor.steps_completed >= 2, timeout=10)
    await asyncio.sleep(1.2)          # old timer deadline passes
    assert actor.state.wait_generation >= gen_before
    # exactly one extra cycle from the message; the dead timer added at
    # most the one scheduled wakeup, not a storm
    assert actor.steps_completed <= 3
