"""Consensus merge rules: how parameter values from multiple models combine.

Behavior-parity with the reference's rule set (reference:
lib/quoracle/actions/consensus_rules.ex:18-163,342-441), re-expressed natively.
The embedding comparison is injectable: in production it is the GPU embedding
engine + the fused cosine HIP kernel (quoracle_amd.ops.cosine_sim_matrix);
tests inject deterministic callables.

Rules:
    exact_match          all values identical, else no consensus
    semantic_similarity  embeddings of all values within cosine >= threshold of
                         the first; first value wins
    mode_selection       most frequent value (first-seen wins ties)
    union_merge          flatten + dedupe, order-preserving
    structural_merge     deep dict merge, later values override
    percentile(n)        linear-interpolated percentile of numeric values,
                         rounded to int; falls back to mode for non-numerics
    wait_parameter       median with special boolean handling
    batch_sequence_merge position-wise merge of equal-length action sequences
    first_non_nil        first present value
    merge_maps           shallow dict merge, later values override
"""

from __future__ import annotations

import math
from typing import Any, Callable, Dict, List, Optional, Sequence

from ..utils.jsonx import dumps_canonical

# Batch embedder: list of texts -> list of vectors (anything indexable with
# float elements — list, np.ndarray row, torch tensor row).
EmbedManyFn = Callable[[List[str]], Sequence[Sequence[float]]]


class NoConsensus(Exception):
    """Raised when a rule cannot merge the given values."""

    def __init__(self, reason: str = "no_consensus"):
        super().__init__(reason)
        self.reason = reason


def cosine_similarity(v1: Sequence[float], v2: Sequence[float]) -> float:
    """Plain cosine similarity; zero vectors compare as 0.0.

    CPU reference path — the GPU vote path computes the full similarity
    matrix in one fused HIP kernel instead (ops.cosine_sim_matrix).
    """
    if len(v1) != len(v2):
        raise ValueError("Vectors must have the same length")
    dot = m1 = m2 = 0.0
    for a, b in zip(v1, v2):
        a = float(a)
        b = float(b)
        dot += a * b
        m1 += a * a
        m2 += b * b
    if m1 == 0.0 or m2 == 0.0:
        return 0.0
    return dot / (math.sqrt(m1) * math.sqrt(m2))


def _freeze(value: Any) -> Any:
    """Hashable key for counting arbitrary JSON-ish values."""
    if isinstance(value, (dict, list)):
        return dumps_canonical(value)
    return (type(value).__name__, value)


def mode_value(values: Sequence[Any]) -> Any:
    """Most frequent value; first-encountered wins ties."""
    counts: Dict[Any, int] = {}
    first_at: Dict[Any, int] = {}
    for i, v in enumerate(values):
        k = _freeze(v)
        counts[k] = counts.get(k, 0) + 1
        first_at.setdefault(k, i)
    best = max(counts.items(), key=lambda kv: (kv[1], -first_at[kv[0]]))
    return values[first_at[best[0]]]


def _median_int(sorted_vals: List[float]) -> Any:
    n = len(sorted_vals)
    mid = n // 2
    if n % 2 == 0:
        left, right = sorted_vals[mid - 1], sorted_vals[mid]
        if isinstance(left, int) and isinstance(right, int):
            return (left + right) // 2
        return (left + right) / 2
    return sorted_vals[mid]


def _percentile(sorted_vals: List[float], pct: float) -> int:
    index = pct / 100.0 * (len(sorted_vals) - 1)
    lo = int(index)
    hi = min(lo + 1, len(sorted_vals) - 1)
    frac = index - lo
    result = sorted_vals[lo] + (sorted_vals[hi] - sorted_vals[lo]) * frac
    # Banker's rounding differs from Elixir round/1 (half away from zero);
    # match the reference for .5 cases.
    return int(math.floor(result + 0.5)) if result >= 0 else int(math.ceil(result - 0.5))


def merge_wait_values(values: Sequence[Any]) -> Any:
    """The wait-parameter rule (reference: consensus_rules.ex:114-160)."""
    if not values:
        raise NoConsensus("no_values")
    booleans = [v for v in values if isinstance(v, bool)]
    import math as _math
    integers = [v for v in values if isinstance(v, (int, float))
                and not isinstance(v, bool) and _math.isfinite(v)]
    # drop unusable shapes ("wait": null / NaN / strings); all-unusable -> no wait
    values = booleans + integers
    if not values:
        return False

    if not integers and booleans and all(b is False for b in booleans):
        return False
    if not integers and booleans and all(b is True for b in booleans):
        return True
    if not integers and len(booleans) >= 3 and any(booleans):
        # mixed booleans, 3+ voters, any true -> true
        return True
    if not booleans and integers:
        return _median_int(sorted(integers))
    # Mixed types (or 2 mixed booleans): convert booleans and take median.
    max_int = max(integers) if integers else 30
    converted = [0 if v is False else (max_int if v is True else v) for v in values]
    return _median_int(sorted(converted))


def deep_merge(d1: Any, d2: Any) -> Any:
    if isinstance(d1, dict) and isinstance(d2, dict):
        out = dict(d1)
        for k, v in d2.items():
            out[k] = deep_merge(out[k], v) if k in out else v
        return out
    if isinstance(d2, dict):
        return d2
    if isinstance(d1, dict):
        return d1
    return d2


def apply_rule(
    rule: Any,
    values: Sequence[Any],
    *,
    embed_many: Optional[EmbedManyFn] = None,
) -> Any:
    """Apply a consensus rule to a list of values.

    Returns the merged value or raises NoConsensus.  `embed_many` backs the
    semantic_similarity rule; without it, differing strings can't agree.
    """
    if rule == "batch_sequence_merge":
        return _batch_sequence_merge(list(values), embed_many=embed_many)
    if not values:
        raise NoConsensus("no_values")
    values = list(values)

    if rule == "exact_match":
        if len(values) == 1:
            return values[0]
        uniq = {_freeze(v) for v in values}
        if len(uniq) == 1:
            return values[0]
        raise NoConsensus()

    if isinstance(rule, tuple) and rule[0] == "semantic_similarity":
        threshold = rule[1] if len(rule) > 1 else 0.9
        return _semantic_merge(values, threshold, embed_many)

    if rule == "mode_selection":
        return mode_value(values)

    if rule == "union_merge":
        flat: List[Any] = []
        for v in values:
            flat.extend(v if isinstance(v, list) else [v])
        seen = set()
        out = []
        for v in flat:
            k = _freeze(v)
            if k not in seen:
                seen.add(k)
                out.append(v)
        return out

    if rule == "structural_merge":
        merged: Any = {}
        for v in values:
            merged = deep_merge(merged, v)
        return merged

    if isinstance(rule, tuple) and rule[0] == "percentile":
        numeric = [v for v in values
                   if isinstance(v, (int, float)) and not isinstance(v, bool)
                   and math.isfinite(v)]
        if not numeric:
            return mode_value(values)
        return _percentile(sorted(numeric), rule[1])

    if rule == "wait_parameter":
        return merge_wait_values(values)

    if rule == "first_non_nil":
        for v in values:
            if v is not None:
                return v
        raise NoConsensus("no_values")

    if rule == "merge_maps":
        out: Dict[str, Any] = {}
        for v in values:
            if isinstance(v, dict):
                out.update(v)
        return out

    raise NoConsensus("unknown_rule")


def _semantic_merge(
    values: List[Any], threshold: float, embed_many: Optional[EmbedManyFn]
) -> Any:
    uniq = {_freeze(v) for v in values}
    if len(uniq) == 1:
        return values[0]
    if embed_many is None:
        raise NoConsensus("embedding_failed")
    texts = [v if isinstance(v, str) else dumps_canonical(v) for v in values]
    try:
        vectors = embed_many(texts)
    except Exception:
        raise NoConsensus("embedding_failed") from None
    if vectors is None or len(vectors) != len(values):
        raise NoConsensus("embedding_failed")
    first = vectors[0]
    for vec in vectors[1:]:
        if cosine_similarity(first, vec) < threshold:
            raise NoConsensus()
    return values[0]


def _normalize_action_spec(spec: Any) -> Dict[str, Any]:
    if not isinstance(spec, dict):
        return {"action": None, "params": {}}
    action = spec.get("action")
    params = spec.get("params")
    return {"action": action,
            "params": params if isinstance(params, dict) else {}}


def _batch_sequence_merge(
    sequences: List[Any], *, embed_many: Optional[EmbedManyFn]
) -> List[Any]:
    """Position-wise merge of batch action sequences.

    All sequences must be equal length and agree on the action type at each
    position; params merge per that action's consensus rules.
    (reference: consensus_rules.ex:376-441)
    """
    from ..actions import schema as schema_mod

    sequences = [s for s in sequences if isinstance(s, list)]
    if not sequences:
        return []
    if len(sequences) == 1:
        return sequences[0]
    lengths = {len(s) for s in sequences}
    if len(lengths) > 1:
        raise NoConsensus("sequence_length_mismatch")

    merged: List[Dict[str, Any]] = []
    for position in zip(*sequences):
        specs = [_normalize_action_spec(s) for s in position]
        action_types = {s["action"] for s in specs}
        if len(action_types) != 1:
            raise NoConsensus("sequence_mismatch")
        action = specs[0]["action"]
        sch = schema_mod.try_get_schema(action)
        if sch is None:
            raise NoConsensus("unknown_action")
        all_keys: List[str] = []
        for s in specs:
            for k in s["params"]:
                if k not in all_keys:
                    all_keys.append(k)
        params: Dict[str, Any] = {}
        for key in all_keys:
            vals = [s["params"][key] for s in specs
                    if s["params"].get(key) is not None]
            if not vals:
                continue
            rule = sch.consensus_rules.get(key, "exact_match")
            params[key] = apply_rule(rule, vals, embed_many=embed_many)
        merged.append({"action": action, "params": params})
    return merged


def merge_param(
    action: str,
    param: str,
    values: Sequence[Any],
    *,
    embed_many: Optional[EmbedManyFn] = None,
) -> Any:
    """Merge one parameter's values using its schema rule.

    The top-level 'wait' field always uses the wait_parameter rule.
    """
    from ..actions import schema as schema_mod

    if param == "wait":
        return apply_rule("wait_parameter", values, embed_many=embed_many)
    sch = schema_mod.try_get_schema(action)
    if sch is None:
        raise NoConsensus("unknown_action")
    rule = sch.consensus_rules.get(param)
    if rule is None:
        raise NoConsensus("unknown_param")
    return apply_rule(rule, values, embed_many=embed_many)
