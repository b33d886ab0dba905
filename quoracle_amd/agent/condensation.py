"""Context condensation: evict old history, reflect, keep lessons + state.

Behavior-parity with the reference (reference: lib/quoracle/agent/consensus/
per_model_query/condensation.ex:39-454, token_manager.ex:152-229):
  * reactive condensation at 100% of the model's context window — remove the
    oldest >80% of tokens, run ACE reflection over the removed text
  * model-initiated `condense: N` removes the N oldest entries
  * proactive condensation when the projected output budget would fall below
    the 4096-token floor
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

    text = _discarded_text(discarded)
    try:
        lessons, model_state = await reflect(engine, model_key, text)
        state.context_lessons[model_key] = merge_lessons(
            state.context_lessons.get(model_key, []), lessons, embed_many)
        if model_state:
            state.model_states[model_key] = model_state
        keep = keep + [history_entry(
            "event",
            f"[{len(discarded)} older history entries were condensed; "
            "their lessons and working state are injected separately]")]
    except Exception:
        # Fallback artifact: keep a truncated summary of what was lost
        # (reference: condensation.ex:439-454)
        logger.warning("reflection failed for %s; keeping fallback artifact",
                       model_key)
        artifact = text[:2000]
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
        if not await condense_model_history(state, model_key, engine,
                                            embed_many=embed_many):
            return input_tokens
    return input_tokens_fn()
