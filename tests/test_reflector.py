"""Direct ACE reflector coverage (agent/reflector.py; reference:
lib/quoracle/agent/reflector.ex:23-94): lesson normalization across the
shapes models actually emit, retry-once on garbage, RuntimeError after both
attempts fail (callers then write the condensation artifact)."""

import json

import pytest

from quoracle_amd.agent.reflector import (MIN_REFLECTION_TOKENS,
                                          build_reflection_messages, reflect)
from quoracle_amd.engine.fake import FakeEngine


@pytest.mark.asyncio
async def test_reflect_normalizes_lesson_shapes():
    payload = json.dumps({
        "lessons": [
            "plain string lesson",
            {"text": "dict lesson", "confidence": 3},
            {"content": "content-key lesson"},
            {"weird": True},
        ],
        "state": {"open_items": ["a"]},
    })
    eng = FakeEngine(responses={"m": [payload]})
    lessons, state = await reflect(eng, "m", "history to discard")
    assert [l["text"] for l in lessons][:3] == [
        "plain string lesson", "dict lesson", "content-key lesson"]
    assert lessons[1]["confidence"] == 3
    assert lessons[0]["confidence"] == 1
    assert json.loads(lessons[3]["text"]) == {"weird": True}
    assert state == {"open_items": ["a"]}
    # min output budget honored
    assert eng.calls[0].max_tokens >= MIN_REFLECTION_TOKENS


@pytest.mark.asyncio
async def test_reflect_retries_once_then_succeeds():
    good = json.dumps({"lessons": ["survivor"], "state": None})
    eng = FakeEngine(responses={"m": ["not json at all", good]})
    lessons, state = await reflect(eng, "m", "x")
    assert lessons[0]["text"] == "survivor" and state is None
    assert len(eng.calls) == 2


@pytest.mark.asyncio
async def test_reflect_raises_after_two_failures():
    eng = FakeEngine(responses={"m": ["garbage", "more garbage"]})
    with pytest.raises(RuntimeError, match="reflection_failed"):
        await reflect(eng, "m", "x")


def test_reflection_prompt_contains_history():
    msgs = build_reflection_messages("THE-DISCARDED-TEXT")
    assert msgs[-1]["role"] == "user"
    assert "THE-DISCARDED-TEXT" in msgs[-1]["content"]
    assert "lessons" in msgs[-1]["content"]
