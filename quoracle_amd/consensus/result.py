"""Consensus result formatting: winner selection, parameter merging,
confidence scoring, and tie-breaking.

Behavior-parity with the reference (reference: lib/quoracle/consensus/result.ex
and result/scoring.ex).  Always yields exactly ONE action decision.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Tuple

from ..actions import schema as schema_mod
from . import rules as rules_mod
from .aggregator import Cluster
from .rules import EmbedManyFn, NoConsensus, mode_value

logger = logging.getLogger(__name__)


@dataclass
class ConsensusDecision:
    kind: str              # "consensus" | "forced_decision"
    action: Dict[str, Any]  # {action, params, reasoning, wait}
    confidence: float
    round_num: int = 1
    clusters: Optional[List[Cluster]] = None


def format_result(
    clusters: List[Cluster],
    total_count: int,
    round_num: int,
    *,
    max_refinement_rounds: int = 4,
    embed_many: Optional[EmbedManyFn] = None,
) -> ConsensusDecision:
    """Pick the winning cluster and merge its parameters.

    Majority (>50%) -> "consensus"; otherwise plurality with tie-break ->
    "forced_decision".
    """
    kind, cluster = _find_winner(clusters, total_count)
    action = merge_cluster_params(cluster, embed_many=embed_many)
    confidence = calculate_confidence(
        cluster.count, total_count, round_num, max_refinement_rounds)
    return ConsensusDecision(
        kind=kind, action=action, confidence=confidence,
        round_num=round_num, clusters=clusters)


def _find_winner(clusters: List[Cluster], total_count: int) -> Tuple[str, Cluster]:
    for cluster in clusters:
        if cluster.count > total_count / 2:
            return "consensus", cluster
    max_count = max(c.count for c in clusters)
    tied = [c for c in clusters if c.count == max_count]
    winner = break_tie(tied) if len(tied) > 1 else tied[0]
    return "forced_decision", winner


def calculate_confidence(
    cluster_count: int, total_count: int, round_num: int,
    max_refinement_rounds: int = 4,
) -> float:
    """proportion + majority bonus − late-round penalty, clamped [0.1, 1.0]."""
    base = cluster_count / total_count
    if base > 0.8:
        bonus = 0.15
    elif base > 0.6:
        bonus = 0.10
    elif base > 0.5:
        bonus = 0.05
    else:
        bonus = 0.0
    penalty = (round_num - max_refinement_rounds) * 0.1 \
        if round_num > max_refinement_rounds else 0.0
    return min(1.0, max(0.1, base + bonus - penalty))


def merge_cluster_params(
    cluster: Cluster, *, embed_many: Optional[EmbedManyFn] = None
) -> Dict[str, Any]:
    """Merge a winning cluster's parameters per-rule into one action."""
    action_type = cluster.representative.get("action")
    if action_type == "batch_sync":
        return _merge_batch_sync(cluster, embed_many=embed_many)
    sch = schema_mod.try_get_schema(action_type)
    if sch is None:
        return dict(cluster.representative)

    all_params = [a.get("params") or {} for a in cluster.actions]
    merged: Dict[str, Any] = {}
    for param in sch.all_params:
        values = [p[param] for p in all_params if p.get(param) is not None]
        if not values:
            continue
        rule = sch.consensus_rules.get(param, "mode_selection")
        try:
            merged[param] = rules_mod.apply_rule(rule, values, embed_many=embed_many)
        except NoConsensus:
            merged[param] = mode_value(values)

    reasoning = next(
        (a.get("reasoning") for a in cluster.actions if a.get("reasoning")), "")
    result = {"action": action_type, "params": merged, "reasoning": reasoning}
    return _attach_wait(result, cluster.actions)


def _attach_wait(result: Dict[str, Any], actions: List[Dict[str, Any]]) -> Dict[str, Any]:
    wait_values = [a.get("wait") for a in actions if a.get("wait") is not None]
    if not wait_values:
        logger.warning("All models omitted wait parameter - defaulting to wait: false")
        result["wait"] = False
        return result
    try:
        result["wait"] = rules_mod.merge_wait_values(wait_values)
    except NoConsensus:
        result["wait"] = wait_values[0]
    return result


def _merge_batch_sync(
    cluster: Cluster, *, embed_many: Optional[EmbedManyFn]
) -> Dict[str, Any]:
    sequences = []
    for action in cluster.actions:
        params = action.get("params") or {}
        sequences.append(params.get("actions") or [])
    try:
        merged_actions = rules_mod.apply_rule(
            "batch_sequence_merge", sequences, embed_many=embed_many)
    except NoConsensus as exc:
        if exc.reason == "no_consensus":
            # param-level disagreement -> position-wise mode fallback
            merged_actions = _merge_sequences_with_mode(sequences)
        else:
            raise
    result = {"action": "batch_sync", "params": {"actions": merged_actions}}
    result["reasoning"] = next(
        (a.get("reasoning") for a in cluster.actions if a.get("reasoning")), "")
    return _attach_wait(result, cluster.actions)


def _merge_sequences_with_mode(sequences: List[List[Dict[str, Any]]]) -> List[Dict[str, Any]]:
    max_len = max((len(s) for s in sequences), default=0)
    merged = []
    for idx in range(max_len):
        at_pos = [s[idx] for s in sequences if idx < len(s)]
        if not at_pos:
            continue
        if len(at_pos) == 1:
            spec = at_pos[0]
            merged.append({"action": spec.get("action"),
                           "params": spec.get("params") or {}})
            continue
        action_type = mode_value([s.get("action") for s in at_pos])
        matching = [s for s in at_pos if s.get("action") == action_type]
        all_params = [s.get("params") or {} for s in matching]
        keys: List[str] = []
        for p in all_params:
            for k in p:
                if k not in keys:
                    keys.append(k)
        params = {}
        for key in keys:
            vals = [p[key] for p in all_params if p.get(key) is not None]
            if vals:
                params[key] = mode_value(vals)
        merged.append({"action": action_type, "params": params})
    return merged


# ---------------------------------------------------------------------------
# Tie-breaking (reference: result/scoring.ex)
# ---------------------------------------------------------------------------

def wait_score(value: Any) -> Tuple[int, int]:
    """(true_count, finite_sum): lower = more conservative = wins."""
    if value is True:
        return (0, 0)
    if value is None:
        return (0, 1)
    if value is False or value == 0:
        return (1, 0)
    if isinstance(value, (int, float)) and value > 0:
        return (0, 1 + int(value))
    return (0, 1)


def cluster_wait_score(cluster: Cluster) -> Tuple[int, int]:
    tc = fs = 0
    for action in cluster.actions:
        t, f = wait_score(action.get("wait"))
        tc += t
        fs += f
    return (tc, fs)


def _cluster_priority(cluster: Cluster) -> int:
    rep = cluster.representative
    action = rep.get("action")
    if action in ("batch_sync", "batch_async"):
        specs = (rep.get("params") or {}).get("actions") or []
        if not specs:
            return 999
        return max(schema_mod.get_action_priority(
            s.get("action") if isinstance(s, dict) else "unknown") for s in specs)
    return schema_mod.get_action_priority(action)


def break_tie(tied: List[Cluster]) -> Cluster:
    """2-level chain: lowest action priority, then lowest wait score."""
    if len(tied) == 1:
        return tied[0]
    return min(tied, key=lambda c: (_cluster_priority(c), cluster_wait_score(c)))
