"""Token accounting and condensation split policy.

The reference estimates tokens with a tiktoken Rust NIF (reference:
lib/quoracle/agent/token_manager.ex:19-24); here counts come from the real
tokenizer of the hosted models via the Engine protocol (free — the engine
tokenizes anyway).  Policy parity:
  * condense when history reaches 100% of the model's context window
    (token_manager.ex:152-160)
  * the condensation split discards the oldest entries holding >80% of the
    tokens (token_manager.ex:177-229)
  * N-oldest split for model-requested `condense: N` (token_manager.ex:247-260)
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Tuple

CONDENSE_TRIGGER_RATIO = 1.0
CONDENSE_DISCARD_RATIO = 0.8


def entry_text(entry: Dict[str, Any]) -> str:
    content = entry.get("content")
    if isinstance(content, str):
        return content
    return json.dumps(content, default=str)


def entry_tokens(count_tokens, entry: Dict[str, Any]) -> int:
    """Token count of one history entry, memoized on the entry.  The memo is
    keyed by content length so any in-place rewrite (e.g. condensation's
    shrink_oversized_entries) invalidates it instead of over-reporting
    forever.  Keeps history_tokens O(new entries) per cycle instead of
    O(history)."""
    text = entry_text(entry)
    cached = entry.get("_tokens")
    if isinstance(cached, int) and entry.get("_tokens_len") == len(text):
        return cached
    n = count_tokens(text)
    try:
        entry["_tokens"] = n
        entry["_tokens_len"] = len(text)
    except TypeError:
        pass
    return n


def history_tokens(count_tokens, history: List[Dict[str, Any]]) -> int:
    return sum(entry_tokens(count_tokens, e) for e in history)


def needs_condensation(count_tokens, history: List[Dict[str, Any]],
                       context_limit: int,
                       trigger_ratio: float = CONDENSE_TRIGGER_RATIO) -> bool:
    return history_tokens(count_tokens, history) >= context_limit * trigger_ratio


def split_for_condensation(
    count_tokens, history_newest_first: List[Dict[str, Any]],
    discard_ratio: float = CONDENSE_DISCARD_RATIO,
) -> Tuple[List[Dict[str, Any]], List[Dict[str, Any]]]:
    """Return (keep_newest_first, discard_oldest_first).

    Walks from the oldest entry, moving entries to the discard pile until the
    discarded token share exceeds discard_ratio.  Always keeps at least the
    newest entry.
    """
    total = history_tokens(count_tokens, history_newest_first)
    if total == 0 or len(history_newest_first) <= 1:
        return list(history_newest_first), []
    target = total * discard_ratio
    discarded = 0
    discard: List[Dict[str, Any]] = []
    oldest_first = list(reversed(history_newest_first))
    idx = 0
    while idx < len(oldest_first) - 1 and discarded < target:
        entry = oldest_first[idx]
        discard.append(entry)
        discarded += entry_tokens(count_tokens, entry)
        idx += 1
    keep_oldest_first = oldest_first[idx:]
    return list(reversed(keep_oldest_first)), discard


def split_n_oldest(
    history_newest_first: List[Dict[str, Any]], n: int
) -> Tuple[List[Dict[str, Any]], List[Dict[str, Any]]]:
    """Discard the N oldest entries (keeping at least the newest one)."""
    if n <= 0 or len(history_newest_first) <= 1:
        return list(history_newest_first), []
    n = min(n, len(history_newest_first) - 1)
    keep = history_newest_first[: len(history_newest_first) - n]
    discard_oldest_first = list(reversed(history_newest_first[len(history_newest_first) - n:]))
    return keep, discard_oldest_first
