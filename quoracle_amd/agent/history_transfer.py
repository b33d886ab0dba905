"""Runtime model-pool switching with history transfer.

Behavior parity with the reference (reference: agent/history_transfer.ex:
38-240): when an agent's model pool changes mid-task, pick the SOURCE model
with the largest history, condense it until it fits the SMALLEST context
window among the target models, then seed every new pool member with that
history (plus the source's lessons/state); models present in both pools
keep their own histories untouched.
"""

from __future__ import annotations

import copy
from typing import Callable, Dict, List

from . import condensation as condensation_mod
from . import token_manager as tm


def _history_tokens(count_tokens: Callable[[str], int], history: List[Dict]) -> int:
    return sum(tm.entry_tokens(count_tokens, e) for e in history)


async def transfer_histories(state, new_pool: List[str], engine_for,
                             embed_many=None) -> Dict[str, str]:
    """Mutate state's per-model maps for the new pool.  Returns a report
    {model: source} for observability."""
    old_pool = list(state.model_pool)
    kept = [m for m in new_pool if m in old_pool]
    added = [m for m in new_pool if m not in old_pool]
    report = {m: "kept" for m in kept}

    if added:
        # source = largest history among the old pool
        source = max(old_pool, key=lambda m: len(state.model_histories.get(m, [])),
                     default=None)
        if source is not None and state.model_histories.get(source):
            src_engine = engine_for(source)
            # shrink a COPY of the source history until it fits the
            # smallest target context window
            smallest = min(added, key=lambda m: engine_for(m).context_limit(m))
            tgt_engine = engine_for(smallest)
            limit = int(tgt_engine.context_limit(smallest) * 0.5)
            history = copy.deepcopy(state.model_histories[source])
            guard = 0
            while (_history_tokens(src_engine.count_tokens, history) > limit
                   and len(history) > 1 and guard < 64):
                guard += 1
                # borrow the source model's slot to run real condensation,
                # then take the shrunken history back out
                saved = state.model_histories[source]
                state.model_histories[source] = history
                try:
                    await condensation_mod.condense_model_history(
                        state, source, src_engine, embed_many=embed_many)
                    history = state.model_histories[source]
                finally:
                    state.model_histories[source] = saved
            lessons = copy.deepcopy(state.context_lessons.get(source, []))
            mstate = copy.deepcopy(state.model_states.get(source))
            for m in added:
                state.model_histories[m] = copy.deepcopy(history)
                state.context_lessons[m] = copy.deepcopy(lessons)
                state.model_states[m] = copy.deepcopy(mstate)
                report[m] = f"seeded_from:{source}"
        else:
            for m in added:
                state.model_histories.setdefault(m, [])
                state.context_lessons.setdefault(m, [])
                state.model_states.setdefault(m, None)
                report[m] = "fresh"

    # drop models not in the new pool
    for m in old_pool:
        if m not in new_pool:
            state.model_histories.pop(m, None)
            state.context_lessons.pop(m, None)
            state.model_states.pop(m, None)
    state.model_pool = list(new_pool)
    state.cached_system_prompt = None      # pool shows up in prompts
    return report
