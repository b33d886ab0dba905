"""Per-model user-message injectors.

The reference injects TODO list, children roster, budget status, ACE
lessons/state, context-size telemetry and correction feedback into the last
user message of each model's conversation — user messages rather than system
so the (KV-cached) system prompt prefix stays byte-stable across cycles
(reference: lib/quoracle/agent/consensus_handler/*_injector.ex, SURVEY.md
§2.1).  Same here: a stable system prompt means the prefix KV pages are
shared across every consensus cycle of an agent.
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Optional


def _append_to_last_user(messages: List[Dict[str, str]], text: str) -> List[Dict[str, str]]:
    """Append a block to the final user message (create one if needed)."""
    if not text:
        return messages
    out = [dict(m) for m in messages]
    for i in range(len(out) - 1, -1, -1):
        if out[i]["role"] == "user":
            out[i]["content"] = out[i]["content"] + "\n\n" + text
            return out
    out.append({"role": "user", "content": text})
    return out


def todo_block(todos: List[Dict[str, Any]]) -> str:
    if not todos:
        return ""
    lines = ["## Your current TODO list"]
    marks = {"todo": "[ ]", "pending": "[~]", "done": "[x]"}
    for item in todos:
        mark = marks.get(item.get("state", "todo"), "[ ]")
        lines.append(f"- {mark} {item.get('content', '')}")
    return "\n".join(lines)


def children_block(children: Dict[str, Dict[str, Any]]) -> str:
    if not children:
        return ""
    lines = ["## Your child agents"]
    for child_id, info in children.items():
        status = info.get("status", "running")
        task = info.get("task_description", "")
        budget = info.get("budget")
        budget_str = f", budget ${budget}" if budget is not None else ""
        lines.append(f"- {child_id} ({status}{budget_str}): {task}")
    return "\n".join(lines)


def budget_block(mode: str, allocated: Optional[float], spent: float,
                 committed: float) -> str:
    if mode == "na":
        return ""
    if allocated is None:
        return "## Budget\nUnlimited budget."
    available = allocated - spent - committed
    ratio = available / allocated if allocated > 0 else 0.0
    if available < 0:
        status = "OVER BUDGET"
    elif ratio <= 0.2:
        status = "WARNING: low budget"
    else:
        status = "ok"
    return ("## Budget\n"
            f"Allocated: ${allocated:.2f} | Spent: ${spent:.2f} | "
            f"Committed to children: ${committed:.2f} | "
            f"Available: ${available:.2f} ({status})")


def ace_block(lessons: List[Dict[str, Any]],
              model_state: Optional[Dict[str, Any]]) -> str:
    parts = []
    if lessons:
        lines = ["## Lessons from condensed context (most recent first)"]
        for lesson in lessons[:20]:
            confidence = lesson.get("confidence", 1)
            lines.append(f"- ({confidence}x) {lesson.get('text', '')}")
        parts.append("\n".join(lines))
    if model_state:
        parts.append("## Working state from condensed context\n"
                     + json.dumps(model_state, indent=2, default=str))
    return "\n\n".join(parts)


def context_telemetry_block(used_tokens: int, context_limit: int) -> str:
    pct = 100.0 * used_tokens / context_limit if context_limit else 0.0
    return (f"## Context usage\n{used_tokens} of {context_limit} tokens "
            f"({pct:.0f}%). You may request condensation of the N oldest "
            f"entries by adding \"condense\": N to your response.")


def correction_block(errors: Dict[str, str], model_key: str) -> str:
    reason = errors.get(model_key)
    if not reason:
        return ""
    return ("## Correction needed\nYour previous response was rejected: "
            f"{reason}. Respond with valid JSON containing 'reasoning', "
            "'action', 'params', and 'wait' fields, using only documented "
            "actions and parameters.")


def inject_all(
    messages: List[Dict[str, str]],
    *,
    todos: Optional[List[Dict[str, Any]]] = None,
    children: Optional[Dict[str, Dict[str, Any]]] = None,
    budget: Optional[Dict[str, Any]] = None,
    lessons: Optional[List[Dict[str, Any]]] = None,
    model_state: Optional[Dict[str, Any]] = None,
    used_tokens: Optional[int] = None,
    context_limit: Optional[int] = None,
    correction: str = "",
    refinement_prompt: Optional[str] = None,
) -> List[Dict[str, str]]:
    blocks = []
    if todos:
        blocks.append(todo_block(todos))
    if children:
        blocks.append(children_block(children))
    if budget:
        blocks.append(budget_block(budget.get("mode", "na"), budget.get("allocated"),
                                   budget.get("spent", 0.0), budget.get("committed", 0.0)))
    ace = ace_block(lessons or [], model_state)
    if ace:
        blocks.append(ace)
    if used_tokens is not None and context_limit:
        blocks.append(context_telemetry_block(used_tokens, context_limit))
    if correction:
        blocks.append(correction)
    out = messages
    combined = "\n\n".join(b for b in blocks if b)
    if combined:
        out = _append_to_last_user(out, combined)
    if refinement_prompt:
        out = list(out) + [{"role": "user", "content": refinement_prompt}]
        from .context import merge_consecutive
        out = merge_consecutive(out)
    return out
