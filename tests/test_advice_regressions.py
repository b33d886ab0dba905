"""Dedicated regressions for the round-1 advisor findings:
each was fixed in round 2 — these pin the behaviors."""

import pytest

from quoracle_amd.engine.fake import FakeEngine

from helpers import IDLE


def test_entry_token_memo_invalidates_on_content_change():
    """a rewritten entry (shrink_oversized_entries) must not
    keep its stale memoized token count."""
    from quoracle_amd.agent import token_manager as tm
    calls = []

    def count(text):
        calls.append(text)
        return len(text.split())

    entry = {"type": "user", "content": "one two three four five"}
    assert tm.entry_tokens(count, entry) == 5
    assert tm.entry_tokens(count, entry) == 5        # memo hit
    assert len(calls) == 1
    entry["content"] = "short"
    assert tm.entry_tokens(count, entry) == 1        # memo invalidated
    assert len(calls) == 2


@pytest.mark.asyncio
async def test_shrink_oversized_entries_drops_memo():
    from quoracle_amd.agent import condensation, token_manager as tm
    from quoracle_amd.agent.state import AgentState
    eng = FakeEngine(default_response=IDLE)
    # fake a tiny context so the big entry is over 25% of the window
    orig = eng.context_limit
    eng.context_limit = lambda key: 200
    state = AgentState(agent_id="a", task_id="t", model_pool=["m"])
    state.init_model_maps()
    big = {"type": "user", "content": "tok " * 400}
    tm.entry_tokens(eng.count_tokens, big)           # memoize the big count
    state.model_histories["m"] = [big]
    changed = await condensation.shrink_oversized_entries(state, "m", eng)
    assert changed
    new_entry = state.model_histories["m"][0]
    n = tm.entry_tokens(eng.count_tokens, new_entry)
    assert n <= 120, f"stale memo survived the rewrite: {n}"


def test_dashboard_has_no_html_interpolation():
    """agent-generated strings must reach the dashboard DOM
    via textContent/createElement, never innerHTML interpolation."""
    import inspect
    from quoracle_amd.ui import server
    src = inspect.getsource(server)
    assert "innerHTML=html" not in src
    assert ".innerHTML=logs" not in src
    assert "textContent" in src


def test_grove_name_traversal_rejected():
    """POST /api/tasks grove names with separators or '..'
    must be rejected before any GROVE.md load."""
    from fastapi.testclient import TestClient
    from quoracle_amd.ui.server import create_app
    from helpers import make_manager
    engine = FakeEngine(default_response=IDLE)
    manager, runtime = make_manager(engine)
    app = create_app(manager)
    with TestClient(app) as client:
        for bad in ("../outside", "a/b", "..", "."):
            r = client.post("/api/tasks", json={"prompt": "x",
                                                "profile": "default",
                                                "grove": bad})
            assert r.status_code == 400, (bad, r.status_code)
            assert "bad_grove" in r.text


def test_summarization_role_model_wins_over_pool(monkeypatch):
    """the configured summarization role model must be used
    even when the parent's pool is non-empty (operator-precedence bug)."""
    import inspect
    from quoracle_amd.agent import supervisor as sup
    src = inspect.getsource(sup)
    # the fixed parenthesization
    assert 'or (state.model_pool[0] if state.model_pool else None)' in src \
        or 'or (state.model_pool[0]\n' in src


def test_grammar_max_tokens_clamped_to_context():
    """the grammar's >=1024 max_tokens bump must be
    re-clamped to the remaining context window."""
    from dataclasses import replace
    from quoracle_amd.engine.api import GenerateRequest
    from quoracle_amd.engine.engine import LocalEngine
    from quoracle_amd.models import config as cfg_mod
    import torch
    # a tiny-context preset so the test runs in seconds on CPU
    cfg_mod.PRESETS["tinyctx"] = replace(cfg_mod.PRESETS["tiny"],
                                         max_context=1024)
    try:
        eng = LocalEngine(["tinyctx#clamp"], device=torch.device("cpu"),
                          kv_blocks_override=512, embed_model_key=None,
                          prefill_chunk=256)
        # prompt within ~120 tokens of max_context: the grammar bump to
        # >=1024 would run positions past the window
        limit = eng.context_limit("tinyctx#clamp")
        prompt = "x" * (limit - 120)
        r = eng.generate_sync(GenerateRequest(
            model_key="tinyctx#clamp",
            messages=[{"role": "user", "content": prompt}],
            temperature=0.5, max_tokens=4096, seed=1,
            action_grammar=True, session_id="clamp"), timeout=300)
        # must not crash past max_context: either a clean result or a
        # structured context error — and if ok, positions stayed in range
        if r.ok:
            assert r.input_tokens + r.output_tokens <= limit
        else:
            assert "context" in (r.error or "")
    finally:
        cfg_mod.PRESETS.pop("tinyctx", None)
