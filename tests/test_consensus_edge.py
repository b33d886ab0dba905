"""Consensus-loop edge semantics through the live agent:
forced decision past max refinement rounds, forced reflection for
single-model pools, wait-timer semantics (wait=N continuation, wait=true
idle-until-message)."""

import asyncio
import time

import pytest

from quoracle_amd.engine.fake import FakeEngine
from quoracle_amd.governance.profiles import Profile

from helpers import IDLE, POOL2, action_json, make_manager, wait_until


@pytest.mark.asyncio
async def test_forced_decision_after_max_rounds():
    """Models never converge: after max_refinement_rounds the most-supported
    action is forced (reference: agent/consensus.ex forced decision)."""
    engine = FakeEngine(response_fn=lambda model, msgs, req: action_json(
        "file_read", {"path": f"/tmp/{model}"}))   # permanent disagreement
    manager, runtime = make_manager(engine)
    decisions = []
    result = await manager.create_task("never agree", "default")
    root_id = result["root_agent_id"]
    runtime.bus.on(f"agents:{root_id}:logs",
                   lambda ev: decisions.append(ev)
                   if ev.type == "consensus_decision" else None)
    ok = await wait_until(lambda: decisions, timeout=10)
    assert ok
    payload = decisions[0].payload
    assert payload["kind"] in ("forced", "forced_decision") or \
        payload.get("rounds", 0) >= 3
    assert payload["confidence"] <= 0.6      # round penalty applied


@pytest.mark.asyncio
async def test_single_model_pool_forces_reflection_round():
    """v39: single-model pools go through a forced self-reflection round
    instead of trivially winning (reference: agent/consensus.ex)."""
    calls = []

    def responder(model, msgs, req):
        calls.append([m.get("content", "") for m in msgs])
        return action_json("wait", {"wait": True}, wait=True)

    engine = FakeEngine(response_fn=responder)
    manager, runtime = make_manager(engine, models=["fake-solo"])
    runtime.profiles.put(Profile(
        name="solo", description="", model_pool=["fake-solo"],
        capability_groups=[], force_reflection=True))
    result = await manager.create_task("think twice", "solo")
    root_id = result["root_agent_id"]
    ok = await wait_until(
        lambda: runtime.registry.lookup(root_id)
        and runtime.registry.lookup(root_id).actor.steps_completed >= 1,
        timeout=10)
    assert ok
    # two generate calls for one decision: initial + reflection round
    assert len(calls) >= 2
    joined = "\n".join(calls[-1])
    assert "review" in joined.lower() or "reconsider" in joined.lower() \
        or "refine" in joined.lower() or "round" in joined.lower()


@pytest.mark.asyncio
async def test_wait_n_seconds_then_continue():
    """wait=N idles the agent for ~N seconds, then a fresh cycle runs
    without any incoming message (reference: wait timer semantics)."""
    seq = [action_json("wait", {"wait": 1}, wait=1),
           action_json("todo", {"items": [{"content": "after-timer",
                                           "state": "todo"}]})]
    state = {"i": 0}

    def responder(model, msgs, req):
        # same response for every model within a round
        idx = min(state["i"] // 2, len(seq) - 1)
        state["i"] += 1
        return seq[idx] if idx < len(seq) else IDLE

    engine = FakeEngine(response_fn=responder)
    manager, runtime = make_manager(engine)
    t0 = time.monotonic()
    result = await manager.create_task("wait then act", "default")
    root_id = result["root_agent_id"]
    ok = await wait_until(
        lambda: runtime.registry.lookup(root_id)
        and runtime.registry.lookup(root_id).actor.state.todos, timeout=10)
    assert ok, "timer never fired a follow-up cycle"
    elapsed = time.monotonic() - t0
    assert elapsed >= 0.9, f"cycle resumed too early ({elapsed:.2f}s)"
    todos = runtime.registry.lookup(root_id).actor.state.todos
    assert todos[0]["content"] == "after-timer"


@pytest.mark.asyncio
async def test_wait_true_idles_until_message():
    engine = FakeEngine(default_response=IDLE)   # wait: true
    manager, runtime = make_manager(engine)
    result = await manager.create_task("idle until poked", "default")
    root_id = result["root_agent_id"]
    actor_ok = await wait_until(
        lambda: runtime.registry.lookup(root_id)
        and runtime.registry.lookup(root_id).actor.steps_completed == 1,
        timeout=10)
    assert actor_ok
    actor = runtime.registry.lookup(root_id).actor
    await asyncio.sleep(0.3)
    assert actor.steps_completed == 1      # no spontaneous cycles
    await manager.send_user_message(result["task_id"], "wake up")
    assert await wait_until(lambda: actor.steps_completed >= 2, timeout=10)
