"""Response clustering for the consensus vote.

Groups model responses by action fingerprint (schema-normalized parameter
signature), finds the winning cluster (round 1: unanimity; rounds 2+: >50%
majority), and builds refinement prompts for further rounds.
Behavior-parity with the reference aggregator (reference:
lib/quoracle/consensus/aggregator.ex:23-103,335-351,129-228).

The GPU analogue of the reference's serial CPU cosine loop lives in
quoracle_amd.ops (fused embedding+cosine vote kernel); this module only
decides WHICH strings need comparing.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple

from ..actions import schema as schema_mod
from ..utils.jsonx import dumps_canonical


@dataclass
class Cluster:
    count: int
    actions: List[Dict[str, Any]]
    representative: Dict[str, Any]
    fingerprint: Any = None


def cluster_responses(responses: Sequence[Dict[str, Any]]) -> List[Cluster]:
    """Cluster parsed responses by action fingerprint, largest first."""
    groups: Dict[str, List[Dict[str, Any]]] = {}
    fingerprints: Dict[str, Any] = {}
    order: List[str] = []
    for resp in responses:
        fp = action_fingerprint(resp)
        key = dumps_canonical(fp)
        if key not in groups:
            groups[key] = []
            fingerprints[key] = fp
            order.append(key)
        groups[key].append(resp)
    clusters = [
        Cluster(count=len(groups[k]), actions=groups[k],
                representative=groups[k][0], fingerprint=fingerprints[k])
        for k in order
    ]
    clusters.sort(key=lambda c: -c.count)
    return clusters


def find_majority_cluster(
    clusters: List[Cluster], total_count: int, round_num: int = 2
) -> Optional[Cluster]:
    """Round 1 requires unanimity so every model sees the others' ideas at
    least once; rounds 2+ require a strict majority."""
    if round_num == 1:
        check = lambda c: c.count == total_count  # noqa: E731
    else:
        check = lambda c: c.count > total_count / 2  # noqa: E731
    for cluster in clusters:
        if check(cluster):
            return cluster
    return None


def _batch_action_types(params: Any) -> List[str]:
    # total over malformed responses: clustering runs on parsed (not yet
    # validated) model output, so params may be any JSON shape
    actions = params.get("actions") if isinstance(params, dict) else None
    if not isinstance(actions, list):
        return []
    out = []
    for spec in actions:
        if isinstance(spec, dict):
            out.append(str(spec.get("action", "unknown")))
        else:
            out.append("unknown")
    return out


def action_fingerprint(response: Dict[str, Any]) -> Tuple[str, Any]:
    """Fingerprint an action for clustering.

    batch_sync clusters by the ordered action-type sequence, batch_async by
    the sorted sequence (order-independent); everything else by a
    schema-normalized param signature.
    """
    action = response.get("action")
    params = response.get("params") or {}
    if not isinstance(params, dict):
        params = {}
    if action == "batch_async":
        return ("batch_async", sorted(_batch_action_types(params)))
    if action == "batch_sync":
        return ("batch_sync", _batch_action_types(params))
    sch = schema_mod.try_get_schema(action)
    if sch is None:
        return (str(action), "invalid")
    return (action, _action_signature(params, sch))


def _action_signature(params: Dict[str, Any], sch: schema_mod.ActionSchema) -> Dict[str, Any]:
    sig: Dict[str, Any] = {}
    for param in sch.all_params:
        if param in params:
            value = params[param]
        else:
            continue
        if value is None:
            continue
        rule = sch.consensus_rules.get(param)
        sig[param] = _normalize_for_signature(value, rule)
    return sig


def _normalize_for_signature(value: Any, rule: Any) -> Any:
    if rule == "exact_match":
        return value
    if isinstance(rule, tuple) and rule[0] == "semantic_similarity":
        return _normalize_semantic_string(value, rule[1])
    if rule == "mode_selection":
        # merged by mode during param merge — never splits clusters
        return "_mode_mergeable"
    if rule == "union_merge":
        return sorted(value, key=dumps_canonical) if isinstance(value, list) else value
    if rule == "structural_merge":
        return _deep_sort(value) if isinstance(value, dict) else value
    if isinstance(rule, tuple) and rule[0] == "percentile":
        # merged numerically during param merge — never splits clusters
        return "_percentile_mergeable"
    return value


_WS_RE = re.compile(r"\s+")
_NONWORD_RE = re.compile(r"[^\w\s]")


def _normalize_semantic_string(value: Any, threshold: float) -> Any:
    """Cheap textual normalization so that near-identical phrasings land in
    the same cluster without an embedding call; the true semantic comparison
    happens during param merge."""
    if not isinstance(value, str):
        return value
    s = value.lower()
    if threshold < 0.95:
        s = _NONWORD_RE.sub("", s)
    s = _WS_RE.sub(" ", s).strip()
    words = sorted(w for w in s.split() if len(w) > 3)[:5]
    return "_".join(words)


def _deep_sort(value: Any) -> Any:
    if isinstance(value, dict):
        return {k: _deep_sort(value[k]) for k in sorted(value)}
    return value


def extract_reasoning_history(previous_rounds: List[List[Dict[str, Any]]]) -> List[List[str]]:
    return [[r.get("reasoning", "") for r in round_responses]
            for round_responses in previous_rounds]


def format_reasoning_history(history: List[List[str]]) -> str:
    lines = []
    for i, round_reasonings in enumerate(history, start=1):
        lines.append(f"Round {i}:")
        for reasoning in round_reasonings:
            if reasoning:
                lines.append(f"  - {reasoning}")
    return "\n".join(lines)


def _actions_as_json(responses: Sequence[Dict[str, Any]]) -> str:
    return "\n\n".join(
        json.dumps(
            {"reasoning": r.get("reasoning", ""), "action": r.get("action"),
             "params": r.get("params", {})},
            indent=2, default=str)
        for r in responses
    )


def build_refinement_prompt(
    responses: Sequence[Dict[str, Any]],
    round_num: int,
    context: Dict[str, Any],
) -> str:
    """Refinement prompt for rounds 2..max: show every proposal's JSON (no
    vote counts) and push the model to critique rather than agree."""
    history = context.get("reasoning_history") or []
    history_text = ""
    if history:
        history_text = ("\n\n**Previous reasoning (all rounds):**\n"
                        + format_reasoning_history(history))
    max_rounds = context.get("max_refinement_rounds", 4)
    final_hint = "\nThis is the final round." if round_num >= max_rounds else ""
    return f"""## Consensus Refinement - Round {round_num}

You are one voice in a multi-model consensus. Several models analyzed this task
independently, each with its own history; their proposals are below.

**Prompt:** {context.get('prompt', '')}

**How this works:**
- Each model keeps independent context and learns from its own interactions.
- Weigh ALL the perspectives, then give your best recommendation.
- There is no answer key — genuine deliberation is the point.
- The recipient of the winning action CANNOT see this discussion, so never
  refer to "Proposal 1/2/3" in your action.

**CRITICAL: Review skeptically — do not just agree.**
- Models agree too easily; resist that.
- For each proposal ask: what is wrong with it, what edge case does it miss,
  what would make it fail? Look for bad assumptions, wrong or missing
  parameters, inefficiency, and security problems.
- Then synthesize an improved action. Put the per-proposal critiques
  (1-4 sentences each) and your synthesis in the "reasoning" field.

**Current proposals (JSON format):**
```json
{_actions_as_json(responses)}
```
{history_text}

Respond with valid JSON only. Restate ALL parameters explicitly — your response
must be completely self-contained.{final_hint}
"""


def build_final_round_prompt(
    responses: Sequence[Dict[str, Any]], context: Dict[str, Any]
) -> str:
    max_rounds = context.get("max_refinement_rounds", 4)
    total_rounds = context.get("total_rounds", max_rounds)
    return f"""## Final Consensus Round

This is the FINAL deliberation round; afterwards the most-supported action runs.

**Prompt:** {context.get('prompt', '')}

**Context:** Models with independent histories have deliberated for
{total_rounds - 1} rounds. The refined proposals:

**Final proposals:**
```json
{_actions_as_json(responses)}
```

**Your task:** Synthesize the correct response. Even now, do not simply side
with the majority — if an unaddressed flaw remains, call it out and fix it.

Respond with valid JSON only: brief synthesis in "reasoning" (1-4 sentences),
then your final action with ALL parameters explicit. Do NOT reference
proposals by number.
"""
