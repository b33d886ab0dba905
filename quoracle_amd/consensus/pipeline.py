"""The consensus pipeline: per-model fan-out -> parse -> validate -> cluster ->
majority or refinement loop -> one decision.

Behavior-parity with the reference's top-level consensus
(reference: lib/quoracle/agent/consensus.ex:64-390, consensus/manager.ex):
  * round 1 needs unanimity, rounds 2+ majority, forced decision past
    max_refinement_rounds (default 4, 0-9 per profile)
  * forced reflection for single-model pools (one refinement round)
  * partial pools tolerated: failed/unparseable models are dropped, their
    errors recorded for correction feedback
  * refinement with sliding window (2 rounds) of reasoning history
  * per-round descending temperature per model family

The fan-out itself is delegated to an async `query_fn` supplied by the agent
(it owns per-model histories, injectors and condensation).  On GPU the
queries for a whole agent tree are continuous-batched by the engine
scheduler; this pipeline just awaits its own requests.
"""

from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from typing import Awaitable, Callable, Dict, List, Optional, Sequence

from ..actions.validator import ValidationError, validate_params
from . import aggregator, parser, result as result_mod
from .result import ConsensusDecision
from .rules import EmbedManyFn

# query_fn(model_key, round_num, refinement_prompt) -> raw text
QueryFn = Callable[[str, int, Optional[str]], Awaitable[Optional[str]]]

DEFAULT_MAX_REFINEMENT_ROUNDS = 4


class ConsensusError(Exception):
    def __init__(self, reason: str, model_errors: Optional[Dict[str, str]] = None):
        super().__init__(reason)
        self.reason = reason
        self.model_errors = model_errors or {}


@dataclass
class ConsensusOutcome:
    decision: ConsensusDecision
    rounds_used: int
    temperatures: Dict[str, float] = field(default_factory=dict)
    model_errors: Dict[str, str] = field(default_factory=dict)
    forced_reflection_applied: bool = False
    elapsed_ms: float = 0.0
    responses_per_round: List[int] = field(default_factory=list)


async def run_consensus(
    model_pool: Sequence[str],
    query_fn: QueryFn,
    *,
    max_refinement_rounds: int = DEFAULT_MAX_REFINEMENT_ROUNDS,
    force_reflection: bool = False,
    embed_many: Optional[EmbedManyFn] = None,
    profile_optional_spawn: bool = False,
    round_num: int = 1,
    prompt: str = "",
    on_round: Optional[Callable[[int, List[dict]], None]] = None,
) -> ConsensusOutcome:
    """Drive the full consensus process and return exactly one decision.

    Raises ConsensusError("all_models_failed" | "all_responses_invalid") when
    no round produces a usable response.
    """
    if not model_pool:
        raise ConsensusError("empty_model_pool")
    started = time.monotonic()
    total = len(model_pool)
    model_errors: Dict[str, str] = {}
    reasoning_history: List[List[dict]] = []  # sliding window of 2 rounds
    responses_per_round: List[int] = []
    forced_reflection_applied = False

    responses = await _query_round(model_pool, query_fn, round_num, None, model_errors)
    valid = _parse_and_validate(responses, model_errors,
                                profile_optional_spawn=profile_optional_spawn)
    if not valid:
        reason = ("all_responses_invalid"
                  if any(v in ("invalid_json", "missing_fields", "unknown_action",
                               "missing_required_param", "unknown_parameter",
                               "invalid_param_type", "invalid_enum_value")
                         for v in model_errors.values())
                  else "all_models_failed")
        raise ConsensusError(reason, model_errors)
    responses_per_round.append(len(valid))

    current = valid
    rnd = round_num
    while True:
        if on_round is not None:
            on_round(rnd, current)
        clusters = aggregator.cluster_responses(current)
        forced = force_reflection and total == 1 and rnd == 1
        winner = aggregator.find_majority_cluster(clusters, total, rnd)
        if winner is not None and not forced:
            decision = result_mod.format_result(
                clusters, total, rnd,
                max_refinement_rounds=max_refinement_rounds, embed_many=embed_many)
            break
        if rnd > max_refinement_rounds:
            decision = result_mod.format_result(
                clusters, total, rnd,
                max_refinement_rounds=max_refinement_rounds, embed_many=embed_many)
            break
        if forced:
            forced_reflection_applied = True

        # Build the refinement prompt from the PAST rounds' reasoning, then
        # slide the window (reference: consensus.ex:332-390, manager.ex).
        context = {
            "prompt": prompt,
            "reasoning_history": aggregator.extract_reasoning_history(reasoning_history),
            "max_refinement_rounds": max_refinement_rounds,
            "total_rounds": rnd,
        }
        refinement_prompt = aggregator.build_refinement_prompt(current, rnd, context)
        reasoning_history = (reasoning_history + [
            [{"action": r.get("action"), "params": r.get("params", {}),
              "reasoning": r.get("reasoning")} for r in current]
        ])[-2:]

        refined_raw = await _query_round(
            model_pool, query_fn, rnd + 1, refinement_prompt, model_errors)
        refined = _parse_and_validate(refined_raw, model_errors,
                                      profile_optional_spawn=profile_optional_spawn)
        if not refined:
            # Every refined response failed: fall back to the previous round.
            decision = result_mod.format_result(
                aggregator.cluster_responses(current), len(current), rnd,
                max_refinement_rounds=max_refinement_rounds, embed_many=embed_many)
            break
        responses_per_round.append(len(refined))
        current = refined
        rnd += 1

    from . import temperature as temp_mod
    temperatures = {m: temp_mod.round_temperature(m, rnd, max_refinement_rounds)
                    for m in model_pool}
    return ConsensusOutcome(
        decision=decision,
        rounds_used=rnd,
        temperatures=temperatures,
        model_errors=model_errors,
        forced_reflection_applied=forced_reflection_applied,
        elapsed_ms=(time.monotonic() - started) * 1000.0,
        responses_per_round=responses_per_round,
    )


async def _query_round(
    model_pool: Sequence[str],
    query_fn: QueryFn,
    round_num: int,
    refinement_prompt: Optional[str],
    model_errors: Dict[str, str],
) -> Dict[str, str]:
    """Fan out to every pool model concurrently; drop failures."""
    async def _one(model_key: str):
        try:
            return model_key, await query_fn(model_key, round_num, refinement_prompt)
        except Exception as exc:  # noqa: BLE001 — a failed model drops out of the vote
            return model_key, ("__error__", str(exc))

    results = await asyncio.gather(*[_one(m) for m in model_pool])
    raw: Dict[str, str] = {}
    for model_key, value in results:
        if value is None:
            model_errors[model_key] = "query_failed"
        elif isinstance(value, tuple) and value[0] == "__error__":
            model_errors[model_key] = value[1]
        else:
            raw[model_key] = value
    return raw


def _parse_and_validate(
    raw: Dict[str, str],
    model_errors: Dict[str, str],
    *,
    profile_optional_spawn: bool,
) -> List[dict]:
    pool = parser.parse_pool_responses(raw)
    model_errors.update(pool.errors)
    valid: List[dict] = []
    for response in pool.valid:
        try:
            response["params"] = validate_params(
                response["action"], response["params"],
                profile_optional=profile_optional_spawn)
        except ValidationError as exc:
            model_errors[response.get("model", "?")] = exc.reason
            continue
        valid.append(response)
    return valid
