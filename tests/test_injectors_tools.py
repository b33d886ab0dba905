"""Direct coverage for the per-model message injectors and the
show-prompts tool (reference: lib/quoracle/agent/consensus_handler/*_injector.ex
— blocks land in the last user message so the system-prompt KV prefix stays
byte-stable across cycles)."""

import os
import subprocess
import sys

import pytest

from quoracle_amd.agent import injectors as I

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_blocks_empty_inputs_render_empty():
    assert I.todo_block([]) == ""
    assert I.children_block({}) == ""
    assert I.budget_block("na", None, 0.0, 0.0) == ""
    assert I.ace_block([], None) == ""
    assert I.correction_block({}, "m") == ""


def test_budget_block_thresholds():
    assert "Unlimited" in I.budget_block("allocated", None, 1.0, 0.0)
    ok = I.budget_block("allocated", 10.0, 2.0, 1.0)
    assert "ok" in ok and "$7.00" in ok
    low = I.budget_block("allocated", 10.0, 7.5, 1.0)
    assert "WARNING" in low
    over = I.budget_block("allocated", 10.0, 12.0, 0.0)
    assert "OVER BUDGET" in over


def test_todo_and_children_blocks():
    block = I.todo_block([{"content": "a", "state": "done"},
                          {"content": "b", "state": "todo"}])
    assert "[x] a" in block and "[ ] b" in block
    kids = I.children_block({"c1": {"status": "running", "budget": 3,
                                    "task_description": "dig"}})
    assert "c1" in kids and "budget $3" in kids and "dig" in kids


def test_inject_all_appends_to_last_user_and_keeps_system_stable():
    messages = [{"role": "system", "content": "SYS"},
                {"role": "user", "content": "u1"},
                {"role": "assistant", "content": "a1"},
                {"role": "user", "content": "u2"}]
    out = I.inject_all(messages, todos=[{"content": "x"}],
                       used_tokens=500, context_limit=1000,
                       correction=I.correction_block({"m": "bad json"}, "m"))
    assert out[0]["content"] == "SYS"                 # prefix untouched
    assert out[1]["content"] == "u1"
    assert "TODO" in out[3]["content"] and out[3]["content"].startswith("u2")
    assert "50%" in out[3]["content"]
    assert "bad json" in out[3]["content"]
    # original list untouched
    assert messages[3]["content"] == "u2"


def test_inject_all_refinement_merges_consecutive_users():
    messages = [{"role": "user", "content": "u"}]
    out = I.inject_all(messages, refinement_prompt="refine now")
    assert len(out) == 1 and "refine now" in out[0]["content"]


def test_inject_all_no_user_message_creates_one():
    out = I.inject_all([{"role": "system", "content": "s"}],
                       todos=[{"content": "t"}])
    assert out[-1]["role"] == "user" and "TODO" in out[-1]["content"]


@pytest.mark.parametrize("name", __import__(
    "quoracle_amd.tools.show_prompts", fromlist=["SCENARIOS"]).SCENARIOS)
def test_show_prompts_scenarios_render(name):
    from quoracle_amd.tools.show_prompts import render_scenario
    text = render_scenario(name)
    assert isinstance(text, str) and len(text) > 40


def test_examples_demo_runs_clean():
    """The examples walkthrough must stay executable (it doubles as living
    documentation of the public API)."""
    out = subprocess.run(
        [sys.executable, "examples/demo.py"], cwd=REPO_ROOT,
        capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, out.stderr[-1500:]
    assert "file_write completed" in out.stdout
    assert "demo complete." in out.stdout



def test_show_task_sparse_db(tmp_path):
    """render_task on a task with no agents/logs degrades gracefully."""
    from quoracle_amd.persistence.store import Store
    from quoracle_amd.tools.show_task import render_task
    store = Store(str(tmp_path / "sparse.db"))
    store.save_task({"task_id": "t-empty", "status": "created",
                     "prompt": "bare prompt", "profile": "default"})
    text = render_task(store, "t-empty")
    assert "bare prompt" in text
    missing = render_task(store, "t-nope")
    assert isinstance(missing, str)


def test_event_history_ring_buffer_bounded():
    from quoracle_amd.events import EventBus, LOG_HISTORY_LIMIT
    bus = EventBus()
    for i in range(LOG_HISTORY_LIMIT + 50):
        bus.broadcast("agents:a:logs", "log", {"i": i})
    hist = bus.history("agents:a:logs")
    assert len(hist) == LOG_HISTORY_LIMIT
    assert hist[-1].payload["i"] == LOG_HISTORY_LIMIT + 49
    assert bus.history("agents:a:unknown") == []


def test_examples_demo_governance_runs_clean():
    out = subprocess.run(
        [sys.executable, "examples/demo_governance.py"], cwd=REPO_ROOT,
        capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, out.stderr[-1500:]
    assert "hard_rule_violation" in out.stdout
    assert "confinement_violation" in out.stdout
    assert "confined write landed: True" in out.stdout
    assert "REDACTED marker present: True" in out.stdout
    assert "leaked into logs/history: False" in out.stdout
