"""Tolerant JSON extraction from LLM output.

Mirrors the behavior of the reference's Utils.JsonExtractor +
Consensus.ActionParser (reference: lib/quoracle/consensus/action_parser.ex):
handles markdown code fences, leading/trailing prose, and picks the first
top-level JSON object found.
"""

from __future__ import annotations

import json
import re
from typing import Any, Optional

_FENCE_RE = re.compile(r"```(?:json)?\s*(.*?)```", re.DOTALL)


def extract_json(text: str) -> Optional[dict]:
    """Extract the first JSON object from raw LLM text.

    Tries, in order: whole string, fenced blocks, first balanced {...} span.
    Returns None when no parseable object is found.
    """
    if not isinstance(text, str):
        return None
    stripped = text.strip()
    obj = _try_parse(stripped)
    if obj is not None:
        return obj
    for match in _FENCE_RE.finditer(text):
        obj = _try_parse(match.group(1).strip())
        if obj is not None:
            return obj
    span = _first_balanced_object(text)
    if span is not None:
        return _try_parse(span)
    return None


def _try_parse(text: str) -> Optional[dict]:
    if not text.startswith("{"):
        return None
    try:
        obj = json.loads(text)
    except (json.JSONDecodeError, ValueError):
        return None
    return obj if isinstance(obj, dict) else None


def _first_balanced_object(text: str) -> Optional[str]:
    start = text.find("{")
    while start != -1:
        depth = 0
        in_str = False
        escape = False
        for i in range(start, len(text)):
            ch = text[i]
            if in_str:
                if escape:
                    escape = False
                elif ch == "\\":
                    escape = True
                elif ch == '"':
                    in_str = False
                continue
            if ch == '"':
                in_str = True
            elif ch == "{":
                depth += 1
            elif ch == "}":
                depth -= 1
                if depth == 0:
                    return text[start : i + 1]
        start = text.find("{", start + 1)
    return None


def dumps_canonical(value: Any) -> str:
    """Deterministic JSON used for fingerprints and hashing."""
    return json.dumps(value, sort_keys=True, separators=(",", ":"), default=str)
