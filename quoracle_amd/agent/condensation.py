"""Context condensation: evict old history, reflect, keep lessons + state.

Behavior-parity with the reference (reference: lib/quoracle/agent/consensus/
per_model_query/condensation.ex:39-454, token_manager.ex:152-229):
  * reactive condensation at 100% of the model's context window — remove the
    oldest >80% of tokens, run ACE reflection over the removed text
  * model-initiated `condense: N` removes the N oldest entries
  * proactive condensation when the projected output budget would fall below
    the 4096-token floor
  * reflection BATCHED by context budget: discarded text is reflected in
    chunks that each fit ~40% of the model's window, lessons/state merged
    progressively (reference: condensation.ex:174-205)
  * oversized single entries are recursively summarized with boundary-aware
    splitting before they can blow the window (reference:
    condensation.ex:262-400)
  * reflector failure leaves a fallback condensation artifact instead

On MI355X the eviction is also the KV-page eviction point: history that
leaves the logical context releases its paged KV blocks, and the next prefill
only re-processes the kept prefix.
"""

from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional, Tuple

from ..engine.api import (MIN_OUTPUT_TOKENS, TOKEN_SAFETY_MARGIN, Engine)
from . import token_manager
from .reflector import reflect
from .lessons import merge_lessons
from .state import AgentState, history_entry

logger = logging.getLogger(__name__)


def _discarded_text(discarded_oldest_first: List[Dict[str, Any]]) -> str:
    return "\n\n".join(token_manager.entry_text(e) for e in discarded_oldest_first)


def _reflection_batches(count_tokens, discarded: List[Dict[str, Any]],
                        budget_tokens: int) -> List[str]:
    """Pack discarded entries (oldest first) into reflection chunks that
    each fit the budget (reference: condensation.ex:174-205)."""
    batches: List[str] = []
    current: List[str] = []
    used = 0
    for entry in discarded:
        text = token_manager.entry_text(entry)
        n = count_tokens(text)
        if current and used + n > budget_tokens:
            batches.append("\n\n".join(current))
            current, used = [], 0
        # a single entry larger than the budget rides alone (the reflector
        # prompt truncates; oversized entries are summarized separately)
        current.append(text)
        used += n
    if current:
        batches.append("\n\n".join(current))
    return batches


def _split_at_boundary(text: str) -> Tuple[str, str]:
    """Split near the midpoint at the best semantic boundary available:
    blank line, then newline, then sentence end, then hard split."""
    mid = len(text) // 2
    for sep in ("\n\n", "\n", ". "):
        left = text.rfind(sep, 0, mid)
        right = text.find(sep, mid)
        cut = left if left != -1 else right
        if right != -1 and (cut == -1 or right - mid < mid - cut):
            cut = right
        if cut != -1 and 0 < cut < len(text) - 1:
            return text[:cut + len(sep)], text[cut + len(sep):]
    return text[:mid], text[mid:]


async def summarize_text(engine: Engine, model_key: str, text: str,
                         budget_tokens: int, depth: int = 0) -> str:
    """Recursively summarize text down to the budget (reference:
    condensation.ex:262-400): split at semantic boundaries, summarize the
    halves, then summarize the joined summaries.  Truncation is the
    fail-safe when the model can't help."""
    if engine.count_tokens(text) <= budget_tokens:
        return text
    if depth >= 4:
        return text[: budget_tokens * 2]
    half_budget = max(128, budget_tokens // 2)
    left, right = _split_at_boundary(text)
    left_s = await summarize_text(engine, model_key, left, half_budget,
                                  depth + 1)
    right_s = await summarize_text(engine, model_key, right, half_budget,
                                   depth + 1)
    joined = left_s + "\n" + right_s
    if engine.count_tokens(joined) <= budget_tokens:
        return joined
    from ..engine.api import GenerateRequest
    try:
        result = await engine.generate(GenerateRequest(
            model_key=model_key,
            messages=[{"role": "user",
                       "content": "Condense the following working notes, "
                                  "keeping every concrete fact and decision:"
                                  "\n\n" + joined}],
            temperature=0.3, max_tokens=min(2048, budget_tokens)))
        if result.ok and result.text.strip():
            out = result.text.strip()
            if engine.count_tokens(out) <= budget_tokens * 2:
                return out
    except Exception:  # noqa: BLE001 — summarization is best-effort
        pass
    return joined[: budget_tokens * 2]


async def shrink_oversized_entries(state: AgentState, model_key: str,
                                   engine: Engine,
                                   max_fraction: float = 0.25) -> bool:
    """Replace any single history entry above max_fraction of the context
    window with its recursive summary."""
    limit = engine.context_limit(model_key)
    budget = int(limit * max_fraction)
    changed = False
    history = state.model_histories.get(model_key, [])
    for i, entry in enumerate(history):
        content = entry.get("content")
        if not isinstance(content, str):
            continue
        if engine.count_tokens(content) <= budget:
            continue
        summary = await summarize_text(engine, model_key, content, budget)
        new_entry = {**entry, "content":
                     "[oversized entry summarized]\n" + summary}
        # content changed: the memoized token count (token_manager.entry_tokens)
        # would otherwise carry the stale oversized figure forever
        new_entry.pop("_tokens", None)
        history[i] = new_entry
        changed = True
    return changed


async def condense_model_history(
    state: AgentState,
    model_key: str,
    engine: Engine,
    *,
    n_oldest: Optional[int] = None,
    embed_many=None,
) -> bool:
    """Condense one model's history in place.  Returns True if anything was
    evicted."""
    history = state.model_histories.get(model_key, [])
    if len(history) <= 1:
        return False
    if n_oldest is not None:
        keep, discarded = token_manager.split_n_oldest(history, n_oldest)
    else:
        keep, discarded = token_manager.split_for_condensation(
            engine.count_tokens, history)
    if not discarded:
        return False

    # reflection batched by context budget (~40% of the window per chunk)
    budget = max(512, int(engine.context_limit(model_key) * 0.4))
    batches = _reflection_batches(engine.count_tokens, discarded, budget)
    try:
        for text in batches:
            lessons, model_state = await reflect(engine, model_key, text)
            state.context_lessons[model_key] = merge_lessons(
                state.context_lessons.get(model_key, []), lessons, embed_many)
            if model_state:
                state.model_states[model_key] = model_state
        keep = keep + [history_entry(
            "event",
            f"[{len(discarded)} older history entries were condensed in "
            f"{len(batches)} reflection batch(es); their lessons and "
            "working state are injected separately]")]
    except Exception:
        # Fallback artifact: keep a truncated summary of what was lost
        # (reference: condensation.ex:439-454)
        logger.warning("reflection failed for %s; keeping fallback artifact",
                       model_key)
        artifact = _discarded_text(discarded)[:2000]
        keep = keep + [history_entry(
            "event",
            "[Condensation fallback artifact — reflection unavailable]\n" + artifact)]
    state.model_histories[model_key] = keep
    return True


async def ensure_fits(
    state: AgentState,
    model_key: str,
    engine: Engine,
    input_tokens_fn,
    *,
    embed_many=None,
    max_passes: int = 3,
) -> int:
    """Proactive condensation loop (reference: per_model_query.ex:149-196):
    while the projected output budget is under the floor, condense.  Returns
    the final input token count."""
    limit = engine.context_limit(model_key)
    for _ in range(max_passes):
        input_tokens = input_tokens_fn()
        budget = limit - int(input_tokens * TOKEN_SAFETY_MARGIN)
        if budget >= MIN_OUTPUT_TOKENS:
            return input_tokens
        condensed = await condense_model_history(state, model_key, engine,
                                                 embed_many=embed_many)
        shrunk = await shrink_oversized_entries(state, model_key, engine)
        if not condensed and not shrunk:
            return input_tokens
    return input_tokens_fn()
