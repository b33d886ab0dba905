"""ACE condensation behavior: 80%-oldest eviction with
reflection lessons, model-initiated condense:N, fallback artifact, lesson
dedup by cosine, proactive ensure_fits loop."""

import json

import pytest

from quoracle_amd.agent import condensation as cond
from quoracle_amd.agent.lessons import merge_lessons
from quoracle_amd.agent.state import AgentState, history_entry
from quoracle_amd.engine.fake import FakeEngine, deterministic_embedding


def _state(model="fake-a", entries=12):
    st = AgentState(agent_id="a", task_id="t", parent_id=None,
                    profile="p", model_pool=[model], capability_groups=[])
    st.init_model_maps()
    for i in range(entries):
        st.append_history(history_entry("event", f"step {i} " + "x" * 50))
    return st


def _reflective_engine():
    # reflector asks the SAME model to extract lessons/state as JSON
    return FakeEngine(default_response=json.dumps({
        "lessons": [{"text": "always check the logs", "confidence": 1}],
        "state": {"progress": "mid-task"}}))


@pytest.mark.asyncio
async def test_condensation_evicts_and_extracts_lessons():
    st = _state()
    eng = _reflective_engine()
    before = len(st.model_histories["fake-a"])
    ok = await cond.condense_model_history(st, "fake-a", eng)
    assert ok
    after = st.model_histories["fake-a"]
    assert len(after) < before
    # eviction marker appended at the oldest end
    assert "condensed" in after[-1]["content"]
    lessons = st.context_lessons["fake-a"]
    assert lessons and lessons[0]["text"] == "always check the logs"
    assert st.model_states["fake-a"] == {"progress": "mid-task"}


@pytest.mark.asyncio
async def test_model_initiated_condense_n():
    st = _state(entries=10)
    eng = _reflective_engine()
    await cond.condense_model_history(st, "fake-a", eng, n_oldest=4)
    # 10 - 4 evicted + 1 marker
    assert len(st.model_histories["fake-a"]) == 7


@pytest.mark.asyncio
async def test_reflection_failure_keeps_fallback_artifact():
    st = _state()
    eng = FakeEngine(default_response="not json at all {{{")
    eng.fail_model("fake-a")
    ok = await cond.condense_model_history(st, "fake-a", eng)
    assert ok
    tail = st.model_histories["fake-a"][-1]["content"]
    assert "fallback artifact" in tail
    assert "step 0" in tail          # the lost content survives truncated


@pytest.mark.asyncio
async def test_ensure_fits_condenses_until_output_floor():
    st = _state(entries=30)
    eng = _reflective_engine()
    # context limit so small that the projected output budget is below the
    # 4096-token floor until history shrinks
    eng._context_limits["fake-a"] = 4300
    calls = {"n": 0}

    def fake_input_tokens():
        calls["n"] += 1
        h = st.model_histories["fake-a"]
        return sum(eng.count_tokens(str(e.get("content", ""))) for e in h) + 100

    await cond.ensure_fits(st, "fake-a", eng, fake_input_tokens)
    assert calls["n"] >= 2           # at least one condensation pass ran
    assert len(st.model_histories["fake-a"]) < 30


def test_lesson_dedup_cosine_and_prune():
    embed = lambda texts: [deterministic_embedding(t) for t in texts]
    base = [{"text": "always check the logs first", "confidence": 1}]
    merged = merge_lessons(base,
                           [{"text": "always check the logs first",
                             "confidence": 1}], embed)
    # near-identical lesson merges: confidence bumped, no duplicate
    assert len(merged) == 1 and merged[0]["confidence"] >= 2
    merged = merge_lessons(merged,
                           [{"text": "entirely unrelated topic about gpus",
                             "confidence": 1}], embed)
    assert len(merged) == 2
    # prune keeps at most 100
    many = [{"text": f"unique lesson number {i} {'z' * i}",
             "confidence": 1} for i in range(130)]
    pruned = merge_lessons([], many, embed)
    assert len(pruned) <= 100


def test_lesson_dedup_via_fused_similarity_facade():
    """With an engine exposing similarity_matrix (the GPU cosine-kernel
    seam), dedup uses one batched pass — including new-vs-new pairs."""
    from quoracle_amd.engine.fake import FakeEngine
    from quoracle_amd.engine.pool import EnginePool
    pool = EnginePool(default=FakeEngine(), embedder=FakeEngine())
    facade = pool.embed_facade
    assert facade.has_similarity
    base = [{"text": "check the logs before restarting", "confidence": 1}]
    new = [{"text": "check the logs before restarting", "confidence": 1},
           {"text": "a totally different insight about caching", "confidence": 1},
           {"text": "a totally different insight about caching", "confidence": 1}]
    merged = merge_lessons(base, new, facade)
    texts = [l["text"] for l in merged]
    # exact duplicate merged into base; the two identical new ones merged
    assert len(merged) == 2
    assert merged[0]["confidence"] >= 2 or merged[1]["confidence"] >= 2


@pytest.mark.asyncio
async def test_reflection_batched_by_context_budget():
    """Large discarded spans reflect in multiple context-sized batches
    (reference: condensation.ex:174-205)."""
    st = _state(entries=0)
    for i in range(12):
        st.append_history(history_entry("event", f"note {i} " + "x" * 400))
    eng = _reflective_engine()
    eng._context_limits["fake-a"] = 1000   # batch budget ~400 tokens
    ok = await cond.condense_model_history(st, "fake-a", eng)
    assert ok
    reflect_calls = [c for c in eng.calls]
    assert len(reflect_calls) >= 2, "reflection was not batched"
    marker = st.model_histories["fake-a"][-1]["content"]
    assert "reflection batch" in marker


@pytest.mark.asyncio
async def test_oversized_entry_recursively_summarized():
    """A single entry above 25% of the window is replaced by its recursive
    summary (reference: condensation.ex:262-400)."""
    st = _state(entries=1)
    giant = "\n\n".join(f"paragraph {i}: " + "w" * 120 for i in range(40))
    st.append_history(history_entry("event", giant))
    eng = FakeEngine(default_response="condensed summary of the notes")
    eng._context_limits["fake-a"] = 2000
    changed = await cond.shrink_oversized_entries(st, "fake-a", eng)
    assert changed
    texts = [str(e.get("content", ""))
             for e in st.model_histories["fake-a"]]
    big = [t for t in texts if "oversized entry summarized" in t]
    assert big
    assert eng.count_tokens(big[0]) < eng.count_tokens(giant)


def test_ensure_fits_terminates_on_random_histories():
    """Property: for arbitrary history shapes (including single giant
    entries), ensure_fits terminates within its pass budget and only ever
    shrinks the history (reference: per_model_query.ex:149-196)."""
    import asyncio
    import random
    rng = random.Random(1234)
    for trial in range(10):
        n = rng.randint(1, 30)
        history = [history_entry(rng.choice(["user_message", "action_result",
                                             "event"]),
                                 "x" * rng.randint(10, 30_000))
                   for _ in range(n)]
        state = AgentState(agent_id="a", task_id="t", profile="default",
                           model_pool=["m"])
        state.init_model_maps()
        state.model_histories["m"] = list(history)
        engine = FakeEngine(context_limits={"m": 6000})

        def input_tokens():
            return sum(engine.count_tokens(json.dumps(e, default=str))
                       for e in state.model_histories["m"])

        before = input_tokens()
        out = asyncio.run(
            cond.ensure_fits(state, "m", engine, input_tokens))
        after = input_tokens()
        assert after <= before
        assert len(state.model_histories["m"]) >= 1
        assert out == after
