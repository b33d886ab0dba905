"""Parses raw model output into action maps.

Behavior-parity with the reference parser (reference:
lib/quoracle/consensus/action_parser.ex): tolerant JSON extraction, required
action/params/reasoning fields, optional top-level wait (boolean or integer),
optional 'condense' request, unknown-action rejection.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List

from ..actions import schema as schema_mod
from ..utils.jsonx import extract_json


class ParseError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


def parse_response(text: str) -> Dict[str, Any]:
    """Parse one model response into {action, params, reasoning, wait, condense}.

    Raises ParseError with one of: invalid_json, missing_fields, unknown_action.
    """
    parsed = extract_json(text) if isinstance(text, str) else None
    if parsed is None:
        raise ParseError("invalid_json")

    action = parsed.get("action")
    if not isinstance(action, str):
        raise ParseError("missing_fields")
    params = parsed.get("params")
    if not isinstance(params, dict):
        raise ParseError("missing_fields")
    reasoning = parsed.get("reasoning")
    if not isinstance(reasoning, str):
        raise ParseError("missing_fields")
    if action not in schema_mod.ACTIONS:
        raise ParseError("unknown_action")

    wait = parsed.get("wait")
    if not (wait is None or isinstance(wait, bool) or
            (isinstance(wait, (int, float)) and not isinstance(wait, bool))):
        wait = None

    result: Dict[str, Any] = {
        "action": action,
        "params": params,
        "reasoning": reasoning,
        "wait": wait,
    }
    # Model-initiated condensation: "condense": N oldest entries
    condense = parsed.get("condense")
    if isinstance(condense, int) and not isinstance(condense, bool) and condense > 0:
        result["condense"] = condense
    return result


@dataclass
class ParsedPool:
    """Per-model parse outcomes for one consensus round."""
    valid: List[Dict[str, Any]]
    errors: Dict[str, str]  # model_key -> reason


def parse_pool_responses(raw: Dict[str, str]) -> ParsedPool:
    """Parse each model's raw text; failures are dropped (consensus proceeds
    with partial responses) but recorded for correction feedback."""
    valid: List[Dict[str, Any]] = []
    errors: Dict[str, str] = {}
    for model_key, text in raw.items():
        try:
            parsed = parse_response(text)
        except ParseError as exc:
            errors[model_key] = exc.reason
            continue
        parsed["model"] = model_key
        valid.append(parsed)
    return ParsedPool(valid=valid, errors=errors)
