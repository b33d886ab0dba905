"""ACE reflection: a model self-reflects on history it is about to lose,
extracting durable lessons and working state.

Behavior-parity with the reference (reference: lib/quoracle/agent/
reflector.ex:23-94): the SAME model that owns the history does the
reflection; output is JSON {lessons: [...], state: {...}}; one retry with a
minimum output budget.  On GPU this is just another generate() on the
engine — it batches with everything else.
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Optional, Tuple

from ..engine.api import Engine, GenerateRequest
from ..utils.jsonx import extract_json

MIN_REFLECTION_TOKENS = 128

REFLECTION_PROMPT = """You are about to lose the conversation history below to
context condensation. Reflect on it and extract what must survive.

Respond with EXACTLY ONE JSON object:
{
  "lessons": ["durable, generally-applicable lesson", ...],
  "state": { "working state: current facts, open items, partial results": "..." }
}

Keep lessons short and general; put task-specific facts in "state".

History to be discarded (oldest first):
"""


def build_reflection_messages(discarded_text: str) -> List[Dict[str, str]]:
    return [{"role": "user", "content": REFLECTION_PROMPT + discarded_text}]


async def reflect(
    engine: Engine,
    model_key: str,
    discarded_text: str,
    *,
    max_tokens: int = 2048,
) -> Tuple[List[Dict[str, Any]], Optional[Dict[str, Any]]]:
    """Run reflection; returns (lessons, state).

    Raises RuntimeError after the retry also fails — callers fall back to a
    condensation artifact (reference: condensation.ex:439-454).
    """
    request = GenerateRequest(
        model_key=model_key,
        messages=build_reflection_messages(discarded_text),
        temperature=0.3,
        max_tokens=max(max_tokens, MIN_REFLECTION_TOKENS),
    )
    for attempt in range(2):
        result = await engine.generate(request)
        if not result.ok:
            continue
        parsed = extract_json(result.text)
        if parsed is None:
            continue
        lessons_raw = parsed.get("lessons")
        state = parsed.get("state")
        def _text(l):
            if isinstance(l, str):
                return l
            if isinstance(l, dict):
                t = l.get("text") or l.get("content") or l.get("lesson")
                if isinstance(t, str):
                    return t
            return json.dumps(l, default=str)

        def _conf(l):
            c = l.get("confidence") if isinstance(l, dict) else None
            return c if isinstance(c, (int, float)) else 1

        lessons = [{"text": _text(l), "confidence": _conf(l)}
                   for l in (lessons_raw or []) if l]
        return lessons, state if isinstance(state, dict) else None
    raise RuntimeError("reflection_failed")
