"""End-to-end orchestrator tests with scripted engines through the production
path (the reference's tier-2 fake-backend strategy)."""

import asyncio
import json

import pytest

from quoracle_amd.engine.fake import FakeEngine

from helpers import (IDLE, POOL2, action_json, make_manager, make_runtime,
                     wait_until)


@pytest.mark.asyncio
async def test_task_creates_root_agent_and_decides():
    engine = FakeEngine(default_response=IDLE)
    todo = action_json("todo", {"items": [{"content": "step 1", "state": "todo"}]})
    for m in POOL2:
        engine.push_response(m, todo)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("do something", "default")
    root_id = result["root_agent_id"]
    assert runtime.registry.alive(root_id)

    ok = await wait_until(
        lambda: runtime.registry.lookup(root_id) is not None
        and runtime.registry.lookup(root_id).actor.state.todos)
    assert ok, "todo action never executed"
    actor = runtime.registry.lookup(root_id).actor
    assert actor.state.todos[0]["content"] == "step 1"
    # decision + result entries landed in every model's history
    for m in POOL2:
        types = [e["type"] for e in actor.state.model_histories[m]]
        assert "decision" in types
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_refinement_round_on_disagreement():
    engine = FakeEngine(default_response=IDLE)
    # round 1: disagreement on exact-match param -> two clusters
    engine.push_response("fake-a", action_json("file_read", {"path": "/a"}))
    engine.push_response("fake-b", action_json("file_read", {"path": "/b"}))
    # round 2: both converge
    engine.push_response("fake-a", action_json("file_read", {"path": "/tmp/x"}))
    engine.push_response("fake-b", action_json("file_read", {"path": "/tmp/x"}))
    manager, runtime = make_manager(engine)
    result = await manager.create_task("read the file", "default")
    root_id = result["root_agent_id"]

    def got_result():
        entry = runtime.registry.lookup(root_id)
        if entry is None:
            return False
        h = entry.actor.state.model_histories["fake-a"]
        return any(e["type"] == "result" for e in h)

    assert await wait_until(got_result)
    # refinement prompt was sent: at least 4 generate calls before defaults
    refinement_calls = [c for c in engine.calls
                        if any("Consensus Refinement" in m["content"]
                               for m in c.messages)]
    assert refinement_calls, "no refinement round happened"
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_round1_unanimity_required():
    """Two models agreeing on action but differing on an exact-match param
    must NOT settle in round 1."""
    engine = FakeEngine(default_response=IDLE)
    engine.push_response("fake-a", action_json("send_message",
                                               {"to": "parent", "content": "hello"}))
    engine.push_response("fake-b", action_json("send_message",
                                               {"to": "children", "content": "hello"}))
    engine.push_response("fake-a", action_json("send_message",
                                               {"to": "parent", "content": "hello"}))
    engine.push_response("fake-b", action_json("send_message",
                                               {"to": "parent", "content": "hello"}))
    manager, runtime = make_manager(engine)
    result = await manager.create_task("report", "default")
    root_id = result["root_agent_id"]

    # root send_message to parent routes to the user mailbox topic
    q = runtime.bus.subscribe(f"tasks:{result['task_id']}:messages")
    event = await asyncio.wait_for(q.get(), timeout=5)
    assert event.payload["to"] == "user"
    assert event.payload["content"] == "hello"
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_spawn_child_and_message_flow():
    engine = FakeEngine(default_response=IDLE)
    spawn = action_json("spawn_child", {
        "task_description": "analyze the data",
        "success_criteria": "a summary exists",
        "immediate_context": "data is in /tmp",
        "approach_guidance": "be quick",
        "profile": "default",
        "budget": "10.00",
    }, wait=True)
    for m in POOL2:
        engine.push_response(m, spawn)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("parent task", "default",
                                       budget_limit=100.0)
    root_id = result["root_agent_id"]

    def child_running():
        entry = runtime.registry.lookup(root_id)
        if not entry:
            return False
        kids = runtime.registry.children_of(root_id)
        return bool(kids) and all(
            runtime.registry.lookup(k) is not None for k in kids)

    assert await wait_until(child_running)
    root = runtime.registry.lookup(root_id).actor
    kids = runtime.registry.children_of(root_id)
    assert len(kids) == 1
    child = runtime.registry.lookup(kids[0]).actor
    # escrow locked on parent
    assert root.state.budget_committed == pytest.approx(10.0)
    assert child.state.budget_allocated == pytest.approx(10.0)
    assert child.state.profile == "default"
    # child got the initial task message in history eventually
    assert await wait_until(lambda: any(
        "analyze the data" in str(e.get("content", ""))
        for e in child.state.model_histories["fake-a"]))
    # child_spawned notification reached the parent history
    assert await wait_until(lambda: any(
        "is now running" in str(e.get("content", ""))
        for e in root.state.model_histories["fake-a"]))
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_dismiss_child_releases_escrow_and_terminates_tree():
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("t", "default", budget_limit=100.0)
    root_id = result["root_agent_id"]
    root = runtime.registry.lookup(root_id).actor

    spawn_result = await runtime.supervisor.spawn_child_action(root, {
        "task_description": "x", "success_criteria": "y",
        "immediate_context": "z", "approach_guidance": "w",
        "profile": "default", "budget": "20.00"})
    child_id = spawn_result["child_id"]
    assert await wait_until(lambda: runtime.registry.alive(child_id))
    assert root.state.budget_committed == pytest.approx(20.0)

    await runtime.supervisor.dismiss_child_action(root, child_id, "done")
    assert await wait_until(lambda: not runtime.registry.alive(child_id))
    assert await wait_until(
        lambda: root.state.budget_committed == pytest.approx(0.0))
    assert child_id not in root.state.children
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_consensus_stall_notifies_user():
    engine = FakeEngine(default_response="THIS IS NOT JSON")
    manager, runtime = make_manager(engine)
    result = await manager.create_task("t", "default")
    q = runtime.bus.subscribe(f"tasks:{result['task_id']}:messages")
    event = await asyncio.wait_for(q.get(), timeout=10)
    assert "stalled" in event.payload["content"]
    await manager.supervisor.terminate_tree(result["root_agent_id"])


@pytest.mark.asyncio
async def test_action_gate_blocks_uncapable_action():
    from quoracle_amd.governance.profiles import Profile
    engine = FakeEngine(default_response=IDLE)
    shell = action_json("execute_shell", {"command": "echo hi"})
    for m in POOL2:
        engine.push_response(m, shell)
    manager, runtime = make_manager(engine)
    runtime.profiles.put(Profile(name="restricted", model_pool=POOL2,
                                 capability_groups=[]))  # no local_execution
    result = await manager.create_task("t", "restricted")
    root_id = result["root_agent_id"]
    root = runtime.registry.lookup(root_id).actor
    assert await wait_until(lambda: any(
        "action_not_allowed" in str(e.get("content", ""))
        for e in root.state.model_histories["fake-a"]))
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_pause_and_restore_task():
    engine = FakeEngine(default_response=IDLE)
    todo = action_json("todo", {"items": [{"content": "persist me",
                                           "state": "pending"}]})
    for m in POOL2:
        engine.push_response(m, todo)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("t", "default")
    task_id, root_id = result["task_id"], result["root_agent_id"]
    root = runtime.registry.lookup(root_id).actor
    assert await wait_until(lambda: root.state.todos)

    await manager.pause_task(task_id)
    assert not runtime.registry.alive(root_id)
    assert runtime.store.get_task(task_id)["status"] == "paused"

    restore = await manager.restore_task(task_id)
    assert root_id in restore["restored"]
    restored = runtime.registry.lookup(root_id).actor
    assert restored.state.todos[0]["content"] == "persist me"
    # history survived the round trip
    assert any(e["type"] == "decision"
               for e in restored.state.model_histories["fake-a"])
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_boot_revival():
    engine = FakeEngine(default_response=IDLE)
    from quoracle_amd.persistence.store import Store
    store = Store(":memory:")
    manager, runtime = make_manager(engine, store=store)
    result = await manager.create_task("t", "default")
    task_id, root_id = result["task_id"], result["root_agent_id"]
    assert await wait_until(lambda: runtime.registry.lookup(root_id) is not None
                            and runtime.registry.lookup(root_id).actor.steps_completed >= 0)
    # hard "crash": drop the runtime without pausing
    entry = runtime.registry.lookup(root_id)
    entry.actor._running = False
    entry.actor._task.cancel()
    runtime.registry.unregister(root_id)

    # new runtime over the same store
    manager2, runtime2 = make_manager(FakeEngine(default_response=IDLE),
                                      store=store)
    results = await manager2.restore_running_tasks()
    assert root_id in results[task_id]["restored"]
    assert runtime2.registry.alive(root_id)
    await manager2.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_history_transfer_on_pool_switch():
    """Pool switch seeds new models from the largest source history and
    drops removed models (reference: agent/history_transfer.ex)."""
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("transfer me", "default")
    root = runtime.registry.lookup(result["root_agent_id"]).actor

    from quoracle_amd.agent.state import history_entry
    for i in range(5):
        root.state.append_history(history_entry("event", f"note {i}"),
                                  models=["fake-a"])
    root.state.context_lessons["fake-a"] = [{"content": "lesson", "confidence": 1}]

    report = await root.switch_model_pool(["fake-a", "fake-x"])
    assert report["fake-a"] == "kept"
    assert report["fake-x"].startswith("seeded_from:fake-a")
    assert root.state.model_pool == ["fake-a", "fake-x"]
    assert "fake-b" not in root.state.model_histories
    assert len(root.state.model_histories["fake-x"]) >= 5
    assert root.state.context_lessons["fake-x"] == \
        root.state.context_lessons["fake-a"]
    # deep copies: mutating the clone must not touch the source
    root.state.model_histories["fake-x"][0]["content"] = "mutated"
    assert root.state.model_histories["fake-a"][0]["content"] != "mutated"
    await manager.supervisor.terminate_tree(root.state.agent_id)


@pytest.mark.asyncio
async def test_prompt_tracing_broadcasts_full_exchanges():
    """trace_prompts=True broadcasts every sent message list + raw response
    (reference: consensus_handler.ex debug broadcasts)."""
    from quoracle_amd.tasks.runtime import RuntimeConfig
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine,
                                    config=RuntimeConfig(trace_prompts=True))
    traces = []
    result = await manager.create_task("trace me", "default")
    root = result["root_agent_id"]
    runtime.bus.on(f"agents:{root}:trace", lambda ev: traces.append(ev))
    await manager.send_user_message(result["task_id"], "another turn")
    ok = await wait_until(lambda: len(traces) >= 2)
    assert ok
    ev = traces[0]
    assert ev.payload["model"] in POOL2
    assert ev.payload["messages"][0]["role"] == "system"
    assert isinstance(ev.payload["response"], str)
    await manager.supervisor.terminate_tree(root)


@pytest.mark.asyncio
async def test_child_initial_message_carries_lineage_context():
    """Spawned children receive the parent's summarized decision trail
    (reference: spawn/config_builder.ex ancestor narrative)."""
    engine = FakeEngine(default_response=IDLE)
    spawn = action_json("spawn_child", {
        "task_description": "sub work", "success_criteria": "done",
        "immediate_context": "ctx", "approach_guidance": "go",
        "profile": "default"})
    for m in POOL2:
        engine.push_response(m, action_json(
            "orient", {"current_situation": "investigating the outage",
                       "goal_clarity": "clear",
                       "available_resources": "logs",
                       "key_challenges": "none",
                       "delegation_consideration": "spawn soon"}))
    for m in POOL2:
        engine.push_response(m, spawn)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("lineage root task", "default")
    root_id = result["root_agent_id"]
    assert await wait_until(
        lambda: runtime.registry.children_of(root_id), timeout=10)
    child_id = runtime.registry.children_of(root_id)[0]
    child = runtime.registry.lookup(child_id).actor

    def child_got_initial():
        h = child.state.model_histories.get(POOL2[0], [])
        return any("Lineage context" in str(e.get("content", "")) for e in h)
    assert await wait_until(child_got_initial, timeout=10)
    h = child.state.model_histories[POOL2[0]]
    joined = "\n".join(str(e.get("content")) for e in h)
    assert "orient" in joined        # the parent's decision trail came along
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_announcement_reaches_all_descendants():
    """send_message to='announcement' broadcasts a directive to every
    descendant, depth-first (reference: actions/send_message.ex)."""
    from quoracle_amd.actions import router as R
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("broadcast root", "default")
    root_id = result["root_agent_id"]
    root = runtime.registry.lookup(root_id).actor

    # build a depth-2 tree through the real spawn path
    spawn = {"task_description": "sub", "success_criteria": "s",
             "immediate_context": "c", "approach_guidance": "a",
             "profile": "default"}
    r1 = await manager.supervisor.spawn_child_action(root, dict(spawn))
    assert await wait_until(
        lambda: runtime.registry.children_of(root_id), timeout=10)
    child_id = runtime.registry.children_of(root_id)[0]
    child = runtime.registry.lookup(child_id).actor
    r2 = await manager.supervisor.spawn_child_action(child, dict(spawn))
    assert await wait_until(
        lambda: runtime.registry.children_of(child_id), timeout=10)
    grand_id = runtime.registry.children_of(child_id)[0]

    ctx = R.ActionContext(agent=root, runtime=runtime, action_id="a1",
                          action="send_message",
                          params={"to": "announcement",
                                  "content": "all hands: freeze deploys"})
    res = await R.execute_action(ctx)
    assert set(res["delivered_to"]) == {child_id, grand_id}

    def got(agent_id):
        h = runtime.registry.lookup(agent_id).actor.state.model_histories
        return any("freeze deploys" in str(e.get("content", ""))
                   for e in h[POOL2[0]])
    assert await wait_until(lambda: got(child_id) and got(grand_id), timeout=10)
    await manager.supervisor.terminate_tree(root_id)
