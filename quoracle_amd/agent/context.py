"""ContextManager: per-model history entries -> chat message list.

Behavior-parity with the reference (reference:
lib/quoracle/agent/context_manager.ex:22-259): oldest-first conversion,
entry-type -> role mapping, timestamps prepended to user-role content, and
consecutive same-role merging (the alternation guard).
"""

from __future__ import annotations

import json
import time
from typing import Any, Dict, List, Optional

ROLE_BY_TYPE = {
    "prompt": "user",
    "event": "user",
    "result": "user",
    "user": "user",
    "image": "user",
    "decision": "assistant",
    "assistant": "assistant",
}


def _ts_str(ts: Optional[float]) -> str:
    if not isinstance(ts, (int, float)):
        ts = None
    try:
        return time.strftime("%Y-%m-%d %H:%M:%S UTC",
                             time.gmtime(ts or time.time()))
    except (OSError, OverflowError, ValueError):
        # out-of-range timestamp from a corrupted checkpoint
        return time.strftime("%Y-%m-%d %H:%M:%S UTC", time.gmtime())


def _format_content(entry: Dict[str, Any]) -> str:
    type_ = entry.get("type", "user")
    content = entry.get("content")
    if type_ == "event":
        if isinstance(content, dict) and "from" in content:
            return json.dumps(content, default=str)
        if isinstance(content, dict) and "content" in content:
            return str(content["content"])
        return content if isinstance(content, str) else json.dumps(content, default=str)
    if type_ == "decision":
        return json.dumps(content, default=str) if not isinstance(content, str) else content
    if type_ == "user" and isinstance(content, dict) and "content" in content:
        return str(content["content"])
    if isinstance(content, str):
        return content
    return json.dumps(content, default=str)


def format_history_entry(entry: Dict[str, Any]) -> Dict[str, str]:
    type_ = entry.get("type", "user")
    role = ROLE_BY_TYPE.get(type_, "user")
    content = _format_content(entry)
    if role == "user":
        content = f"[{_ts_str(entry.get('ts'))}] {content}"
    return {"role": role, "content": content}


def merge_consecutive(messages: List[Dict[str, str]]) -> List[Dict[str, str]]:
    """Merge consecutive same-role messages to preserve role alternation."""
    out: List[Dict[str, str]] = []
    for msg in messages:
        if out and out[-1]["role"] == msg["role"]:
            out[-1] = {"role": msg["role"],
                       "content": out[-1]["content"] + "\n\n" + msg["content"]}
        else:
            out.append(dict(msg))
    return out


def build_conversation_messages(
    history_newest_first: List[Dict[str, Any]],
    *,
    context_summary: Optional[str] = None,
    additional_context: Optional[List[Dict[str, str]]] = None,
) -> List[Dict[str, str]]:
    messages: List[Dict[str, str]] = []
    if context_summary:
        messages.append({"role": "system",
                         "content": f"Previous context summary: {context_summary}"})
    messages.extend(additional_context or [])
    for entry in reversed(history_newest_first):
        messages.append(format_history_entry(entry))
    return merge_consecutive(messages)
