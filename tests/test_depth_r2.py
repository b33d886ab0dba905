"""Round-2 depth tests for areas that were one-test-deep (VERDICT r1
weak #8): history transfer scenarios, restore conflict resolution, UI
payload shapes, and lesson-prune boundary behavior."""

import asyncio
import json

import pytest

from quoracle_amd.engine.fake import FakeEngine, deterministic_embedding

from helpers import IDLE, make_manager, wait_until


# ---------------------------------------------------------------------------
# history transfer (reference: agent/history_transfer.ex:38-240)
# ---------------------------------------------------------------------------

def _engine_with_limits(limits):
    eng = FakeEngine(default_response=IDLE)
    orig = eng.context_limit
    eng.context_limit = lambda key: limits.get(key, orig(key))
    return eng


@pytest.mark.asyncio
async def test_history_transfer_kept_models_untouched():
    from quoracle_amd.agent.history_transfer import transfer_histories
    from quoracle_amd.agent.state import AgentState
    eng = FakeEngine(default_response=IDLE)
    state = AgentState(agent_id="a", task_id="t", model_pool=["m1", "m2"])
    state.init_model_maps()
    state.model_histories["m1"] = [{"type": "user", "content": "one"}]
    state.model_histories["m2"] = [{"type": "user", "content": "two"},
                                   {"type": "user", "content": "three"}]
    report = await transfer_histories(state, ["m2", "m3"], lambda m: eng)
    assert report["m2"] == "kept"
    assert report["m3"].startswith("seeded_from:")
    # kept model's history is untouched, in place
    assert [e["content"] for e in state.model_histories["m2"]] == \
        ["two", "three"]
    # dropped model is gone from every per-model map
    assert "m1" not in state.model_histories
    assert "m1" not in state.context_lessons
    assert state.model_pool == ["m2", "m3"]


@pytest.mark.asyncio
async def test_history_transfer_seeds_from_largest_source():
    from quoracle_amd.agent.history_transfer import transfer_histories
    from quoracle_amd.agent.state import AgentState
    eng = FakeEngine(default_response=IDLE)
    state = AgentState(agent_id="a", task_id="t", model_pool=["small", "big"])
    state.init_model_maps()
    state.model_histories["small"] = [{"type": "user", "content": "s"}]
    state.model_histories["big"] = [
        {"type": "user", "content": f"entry {i}"} for i in range(5)]
    state.context_lessons["big"] = [{"text": "lesson from big",
                                     "confidence": 2}]
    state.model_states["big"] = {"note": "state"}
    report = await transfer_histories(state, ["fresh1", "fresh2"],
                                      lambda m: eng)
    assert report["fresh1"] == "seeded_from:big"
    assert report["fresh2"] == "seeded_from:big"
    for m in ("fresh1", "fresh2"):
        assert [e["content"] for e in state.model_histories[m]] == \
            [f"entry {i}" for i in range(5)]
        assert state.context_lessons[m][0]["text"] == "lesson from big"
        assert state.model_states[m] == {"note": "state"}
    # deep copies: mutating one target must not leak into the other
    state.model_histories["fresh1"][0]["content"] = "mutated"
    assert state.model_histories["fresh2"][0]["content"] == "entry 0"


@pytest.mark.asyncio
async def test_history_transfer_condenses_to_smallest_target():
    """A big source history is condensed until it fits 50% of the SMALLEST
    target model's context window."""
    from quoracle_amd.agent.history_transfer import transfer_histories
    from quoracle_amd.agent.state import AgentState
    from quoracle_amd.agent import token_manager as tm
    eng = _engine_with_limits({"tiny-ctx": 400})
    state = AgentState(agent_id="a", task_id="t", model_pool=["src"])
    state.init_model_maps()
    state.model_histories["src"] = [
        {"type": "user", "content": "word " * 120} for _ in range(12)]
    await transfer_histories(state, ["tiny-ctx", "src"], lambda m: eng)
    got = tm.history_tokens(eng.count_tokens,
                            state.model_histories["tiny-ctx"])
    assert got <= 200, f"seeded history {got} tokens > half of 400"
    # source model keeps its own (uncondensed) history
    assert len(state.model_histories["src"]) == 12


@pytest.mark.asyncio
async def test_history_transfer_empty_source_gives_fresh_maps():
    from quoracle_amd.agent.history_transfer import transfer_histories
    from quoracle_amd.agent.state import AgentState
    eng = FakeEngine(default_response=IDLE)
    state = AgentState(agent_id="a", task_id="t", model_pool=["m1"])
    state.init_model_maps()
    report = await transfer_histories(state, ["m2"], lambda m: eng)
    assert report["m2"] == "fresh"
    assert state.model_histories["m2"] == []
    assert state.model_pool == ["m2"]


# ---------------------------------------------------------------------------
# restore conflict resolution (reference: task_restorer/conflict_resolver.ex)
# ---------------------------------------------------------------------------

@pytest.mark.asyncio
async def test_restore_running_task_twice_is_idempotent():
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("idem", "default")
    task_id = result["task_id"]
    root_id = result["root_agent_id"]
    await wait_until(lambda: runtime.registry.lookup(root_id) is not None)
    await manager.pause_task(task_id)
    assert runtime.registry.lookup(root_id) is None
    r1 = await manager.restore_task(task_id)
    r2 = await manager.restore_task(task_id)   # second restore: conflicts
    # exactly one live actor for the agent, registry not corrupted
    assert runtime.registry.lookup(root_id) is not None
    ids = runtime.registry.all_ids()
    assert ids.count(root_id) == 1, (r1, r2, ids)
    await manager.supervisor.terminate_tree(root_id)


@pytest.mark.asyncio
async def test_restore_preserves_tree_topology_and_order():
    """Restore starts parents before children (topological) and rebuilds
    parent/child registry edges for a depth-2 tree."""
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("topo", "default")
    task_id = result["task_id"]
    root = runtime.registry.lookup(result["root_agent_id"]).actor
    spawn_params = {"task_description": "t", "success_criteria": "s",
                    "immediate_context": "c", "approach_guidance": "a",
                    "profile": "default"}
    child = await manager.supervisor.spawn_child_action(root, spawn_params)
    await wait_until(lambda: runtime.registry.lookup(child["child_id"])
                     is not None)
    grand = await manager.supervisor.spawn_child_action(
        runtime.registry.lookup(child["child_id"]).actor, spawn_params)
    await wait_until(lambda: runtime.registry.lookup(grand["child_id"])
                     is not None)
    await manager.pause_task(task_id)
    assert runtime.registry.lookup(result["root_agent_id"]) is None
    await manager.restore_task(task_id)
    # all three back, edges intact
    for aid in (result["root_agent_id"], child["child_id"],
                grand["child_id"]):
        assert runtime.registry.lookup(aid) is not None, aid
    assert child["child_id"] in runtime.registry.children_of(
        result["root_agent_id"])
    assert grand["child_id"] in runtime.registry.children_of(
        child["child_id"])
    await manager.supervisor.terminate_tree(result["root_agent_id"])


# ---------------------------------------------------------------------------
# UI payload shapes (reference: §2.8 typed broadcasts / LiveView payloads)
# ---------------------------------------------------------------------------

@pytest.fixture()
def ui_client():
    from fastapi.testclient import TestClient
    from quoracle_amd.ui.server import create_app
    from quoracle_amd.engine.fake import FakeEngine
    from helpers import make_manager
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    app = create_app(manager)
    with TestClient(app) as client:
        yield client, manager, runtime


def test_task_tree_payload_shape(ui_client):
    client, manager, runtime = ui_client
    created = client.post("/api/tasks", json={"prompt": "shape check",
                                              "profile": "default"}).json()
    assert set(created) >= {"task_id", "root_agent_id"}
    tree = client.get(f"/api/tasks/{created['task_id']}/tree").json()
    assert "agents" in tree and isinstance(tree["agents"], list)
    node = tree["agents"][0]
    assert set(node) >= {"agent_id", "status", "parent_id", "alive"}
    assert node["agent_id"] == created["root_agent_id"]
    assert isinstance(node["alive"], bool)


def test_agent_logs_payload_shape(ui_client):
    client, manager, runtime = ui_client
    created = client.post("/api/tasks", json={"prompt": "logs",
                                              "profile": "default"}).json()
    # seed one log row through the store (the actor loop is not driven
    # under TestClient); the endpoint's serialization shape is the target
    runtime.store.save_log(created["root_agent_id"], created["task_id"],
                           "info", "action_test", "shape probe",
                           {"k": "v"})
    logs = client.get(
        f"/api/agents/{created['root_agent_id']}/logs").json()
    assert isinstance(logs, list) and logs
    entry = logs[0]
    assert set(entry) >= {"level", "event_type", "message"}
    assert all(isinstance(entry[k], str)
               for k in ("level", "event_type", "message"))


def test_engine_stats_payload_is_flat_numeric(ui_client):
    client, _, _ = ui_client
    stats = client.get("/api/engine/stats").json()
    assert isinstance(stats, dict)
    for key, val in stats.items():
        assert isinstance(val, (int, float)), (key, val)


# ---------------------------------------------------------------------------
# lesson prune boundary (reference: lesson_manager.ex prune at 100)
# ---------------------------------------------------------------------------

def test_lesson_prune_boundary_and_order():
    from quoracle_amd.agent.lessons import merge_lessons, MAX_LESSONS
    embed = lambda texts: [deterministic_embedding(t) for t in texts]
    assert MAX_LESSONS == 100
    # exactly at the cap: nothing pruned
    base = [{"text": f"distinct lesson {i} {'x' * (i % 7)}", "confidence": 1}
            for i in range(100)]
    merged = merge_lessons(base, [], embed)
    assert len(merged) == 100
    # one over: pruned back to the cap, and the NEW lesson survives
    extra = {"text": "a brand new lesson about quasars", "confidence": 1}
    merged = merge_lessons(base, [extra], embed)
    assert len(merged) == 100
    assert any("quasars" in l["text"] for l in merged)
    # duplicate merge at the cap must not evict anything (no growth)
    dup = {"text": base[0]["text"], "confidence": 1}
    merged2 = merge_lessons(base, [dup], embed)
    assert len(merged2) == 100
    bumped = [l for l in merged2 if l["text"] == base[0]["text"]]
    assert bumped and bumped[0]["confidence"] >= 2
