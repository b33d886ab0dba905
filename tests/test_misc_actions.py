"""Remaining action coverage: secrets lifecycle through
actions, record_cost, adjust_budget (escrow math + child notification),
skills learn/create with grove-local shadowing, generate_images stub,
dismiss cost absorption."""

import asyncio
import json
import os

import pytest

from quoracle_amd.actions import router as R
from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.governance.profiles import Profile

from helpers import IDLE, action_json, make_manager, make_runtime, wait_until
from test_actions_exec import _actor, _ctx


@pytest.mark.asyncio
async def test_generate_secret_then_use_and_scrub():
    runtime = make_runtime()
    actor = _actor(runtime)
    res = await R.execute_action(_ctx(actor, runtime, "generate_secret",
                                      {"name": "db_pass", "length": 24}))
    assert res["status"] == "created"
    assert res["reference"] == "{{SECRET:db_pass}}"
    value = runtime.vault.get("db_pass")
    assert len(value) == 24
    res = await R.execute_action(_ctx(actor, runtime, "search_secrets",
                                      {"search_terms": ["db"]}))
    assert "db_pass" in res["matches"]
    bad = await R.execute_action(_ctx(actor, runtime, "generate_secret",
                                      {"name": "bad name!"}))
    assert bad.get("error")
    # usage audit rows recorded when resolved through an action
    out = await R.execute_action(_ctx(actor, runtime, "execute_shell",
                                      {"command": "echo {{SECRET:db_pass}}"}))
    usage = runtime.store.secret_usage("db_pass")
    assert usage and usage[0]["agent_id"] == actor.state.agent_id


@pytest.mark.asyncio
async def test_record_cost_updates_budget_and_bus():
    runtime = make_runtime()
    actor = _actor(runtime)
    actor.state.budget_mode = "allocated"
    actor.state.budget_allocated = 10.0
    events = []
    runtime.bus.on(f"agents:{actor.state.agent_id}:costs",
                   lambda ev: events.append(ev))
    res = await R.execute_action(_ctx(actor, runtime, "record_cost",
                                      {"amount": "1.25",
                                       "category": "external_api"}))
    assert res["status"] == "recorded"
    assert actor.state.budget_spent == 1.25
    rows = runtime.store.costs_for_agent(actor.state.agent_id)
    assert any(r["category"] == "external_api" for r in rows)
    bad = await R.execute_action(_ctx(actor, runtime, "record_cost",
                                      {"amount": "-3"}))
    assert bad.get("error")


@pytest.mark.asyncio
async def test_adjust_budget_escrow_and_child_notification():
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("budget tree", "default",
                                       budget_limit=20.0)
    root = runtime.registry.lookup(result["root_agent_id"]).actor
    res = await manager.supervisor.spawn_child_action(root, {
        "task_description": "t", "success_criteria": "s",
        "immediate_context": "c", "approach_guidance": "a",
        "profile": "default", "budget": "5"})
    child_id = res["child_id"]
    assert root.state.budget_committed == 5.0
    assert await wait_until(
        lambda: runtime.registry.alive(child_id), timeout=10)
    child = runtime.registry.lookup(child_id).actor
    assert child.state.budget_allocated == 5.0

    res = await R.execute_action(_ctx(root, runtime, "adjust_budget",
                                      {"child_id": child_id,
                                       "new_budget": "8"}))
    assert res["status"] == "adjusted"
    assert root.state.budget_committed == 8.0
    assert child.state.budget_allocated == 8.0
    # decrease below the child's spend is rejected
    child.state.budget_spent = 7.5
    res = await R.execute_action(_ctx(root, runtime, "adjust_budget",
                                      {"child_id": child_id,
                                       "new_budget": "2"}))
    assert res.get("error") == "budget_below_usage"
    # raising beyond the parent's available budget is rejected
    res = await R.execute_action(_ctx(root, runtime, "adjust_budget",
                                      {"child_id": child_id,
                                       "new_budget": "500"}))
    assert res.get("error") == "insufficient_budget"
    await manager.supervisor.terminate_tree(root.state.agent_id)


@pytest.mark.asyncio
async def test_dismiss_absorbs_child_costs_and_releases_escrow():
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    result = await manager.create_task("absorb", "default", budget_limit=20.0)
    root = runtime.registry.lookup(result["root_agent_id"]).actor
    res = await manager.supervisor.spawn_child_action(root, {
        "task_description": "t", "success_criteria": "s",
        "immediate_context": "c", "approach_guidance": "a",
        "profile": "default", "budget": "4"})
    child_id = res["child_id"]
    assert await wait_until(lambda: runtime.registry.alive(child_id),
                            timeout=10)
    runtime.store.save_cost(child_id, result["task_id"], "m", 1.5)
    spent_before = root.state.budget_spent
    out = await manager.supervisor.dismiss_child_action(root, child_id,
                                                        "done")
    assert out["status"] == "dismissing"
    assert await wait_until(
        lambda: child_id not in root.state.children, timeout=10)
    assert not runtime.registry.alive(child_id)
    assert root.state.budget_committed == 0.0          # escrow released
    assert root.state.budget_spent >= spent_before + 1.5   # costs absorbed
    await manager.supervisor.terminate_tree(root.state.agent_id)


@pytest.mark.asyncio
async def test_skills_create_learn_and_grove_shadowing(tmp_path):
    global_dir = tmp_path / "skills"
    grove_dir = tmp_path / "grove" / "skills"
    os.makedirs(global_dir / "greet")
    (global_dir / "greet" / "SKILL.md").write_text(
        "---\nname: greet\ndescription: global greeting\n---\nSay hello.")
    os.makedirs(grove_dir / "greet")
    (grove_dir / "greet" / "SKILL.md").write_text(
        "---\nname: greet\ndescription: grove greeting\n---\nSay ahoy.")

    from quoracle_amd.tasks.runtime import RuntimeConfig
    runtime = make_runtime(config=RuntimeConfig(skills_dir=str(global_dir)))
    grove = {"name": "g", "path": str(tmp_path / "grove"),
             "skills_path": "skills"}
    actor = _actor(runtime, grove=grove)
    res = await R.execute_action(_ctx(actor, runtime, "learn_skills",
                                      {"skills": ["greet", "nope"]}))
    assert res["missing"] == ["nope"]
    # grove-local skill shadows the global one (reference: skills/loader.ex)
    assert "ahoy" in res["skills"][0]["content"]

    res = await R.execute_action(_ctx(actor, runtime, "create_skill", {
        "name": "made-up", "description": "authored by agent",
        "content": "# Steps\nDo the thing."}))
    assert res["status"] == "created" and os.path.exists(res["path"])
    res = await R.execute_action(_ctx(actor, runtime, "learn_skills",
                                      {"skills": ["made-up"],
                                       "permanent": True}))
    assert res["status"] == "learned_permanently"
    assert any(s["name"] == "made-up" for s in actor.state.active_skills)


@pytest.mark.asyncio
async def test_generate_images_without_model_uses_local_renderer():
    """No injected image_fn: the locally-hosted procedural model renders
    a real PNG and the artifact pipeline stores it (utils/imagegen.py)."""
    runtime = make_runtime()
    actor = _actor(runtime)
    res = await R.execute_action(_ctx(actor, runtime, "generate_images",
                                      {"prompt": "a red square"}))
    assert isinstance(res, dict)
    assert res.get("model") == "local-procedural-v0"
    assert res.get("image_artifacts")


@pytest.mark.asyncio
async def test_show_task_transcript(tmp_path):
    """show-task renders the persisted tree, logs, costs and messages."""
    from quoracle_amd.persistence.store import Store
    from quoracle_amd.tools.show_task import render_task
    from helpers import make_manager, wait_until
    store = Store(str(tmp_path / "t.db"))
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine, store=store)
    result = await manager.create_task("transcribe me", "default")
    root_id = result["root_agent_id"]
    assert await wait_until(
        lambda: runtime.registry.lookup(root_id)
        and runtime.registry.lookup(root_id).actor.steps_completed >= 1,
        timeout=10)
    await manager.send_user_message(result["task_id"], "hello transcript")
    await manager.supervisor.terminate_tree(root_id)
    text = render_task(store, result["task_id"])
    assert "transcribe me" in text
    assert root_id in text
    assert "Agent tree" in text and "Messages" in text


def test_store_concurrent_writes_from_threads(tmp_path):
    """The store is shared across the asyncio runtime and engine threads;
    hammer it from 8 threads and verify every row lands (WAL +
    check_same_thread=False + lock)."""
    import threading
    from quoracle_amd.persistence.store import Store
    store = Store(str(tmp_path / "conc.db"))
    errors = []

    def writer(tid):
        try:
            for i in range(50):
                store.save_log(f"agent-{tid}", "task", "info", "evt",
                               f"msg {i}")
                store.save_cost(f"agent-{tid}", "task", "m", 0.01)
        except Exception as exc:  # noqa: BLE001
            errors.append(exc)

    threads = [threading.Thread(target=writer, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors
    for tid in range(8):
        assert len(store.logs_for_agent(f"agent-{tid}", limit=100)) == 50
        rows = store.costs_for_agent(f"agent-{tid}")
        assert len(rows) == 50


def test_cost_rollup_recursive_cte(tmp_path):
    """Descendant costs aggregate through the recursive CTE over parent
    links (reference: costs/aggregator.ex:122-163): root sees the whole
    subtree, a middle node sees only its own branch."""
    from quoracle_amd.persistence.store import Store
    store = Store(str(tmp_path / "cte.db"))
    #   root -> a -> a1
    #        -> b
    store.save_agent("root", "t", None, config={}, state={}, status="running")
    store.save_agent("a", "t", "root", config={}, state={}, status="running")
    store.save_agent("a1", "t", "a", config={}, state={}, status="running")
    store.save_agent("b", "t", "root", config={}, state={}, status="running")
    store.save_cost("root", "t", "m1", 1.0)
    store.save_cost("a", "t", "m1", 2.0)
    store.save_cost("a1", "t", "m2", 4.0)
    store.save_cost("b", "t", "m1", 8.0)
    root_roll = store.cost_rollup("root")
    assert abs(root_roll["total"] - 15.0) < 1e-9
    a_roll = store.cost_rollup("a")
    assert abs(a_roll["total"] - 6.0) < 1e-9
    leaf = store.cost_rollup("a1")
    assert abs(leaf["total"] - 4.0) < 1e-9
