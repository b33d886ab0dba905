"""Round-based sampling temperature schedule.

Consensus rounds descend from creative (max temperature) to near-deterministic
(floor) linearly over max_refinement_rounds.  Behavior-parity with the
reference (reference: lib/quoracle/consensus/temperature.ex), including the
high-temperature family list kept for heterogeneous pools whose model names
follow provider conventions.  In the local engine the per-round temperature is
simply a per-request sampling parameter consumed by the HIP sampling kernel.
"""

from __future__ import annotations

HIGH_TEMP_FAMILIES = ("gpt", "o1", "o3", "o4", "gemini")
MAX_TEMP_HIGH = 2.0
MAX_TEMP_LOW = 1.0
MIN_TEMP_HIGH = 0.4
MIN_TEMP_LOW = 0.2


def model_name(model_spec: str) -> str:
    """'provider:name' -> 'name'; plain names pass through."""
    if not isinstance(model_spec, str):
        return ""
    parts = model_spec.split(":", 1)
    return parts[1] if len(parts) == 2 else parts[0]


def high_temp_family(name: str) -> bool:
    if not isinstance(name, str):
        return False
    lower = name.lower()
    return any(lower.startswith(f) for f in HIGH_TEMP_FAMILIES)


def max_temperature(model_spec: str) -> float:
    if not isinstance(model_spec, str) or not model_spec:
        return MAX_TEMP_LOW
    return MAX_TEMP_HIGH if high_temp_family(model_name(model_spec)) else MAX_TEMP_LOW


def round_temperature(model_spec: str, round_num: int,
                      max_refinement_rounds: int = 4) -> float:
    """Linear descent from max to floor across the configured rounds,
    rounded to one decimal."""
    max_t = max_temperature(model_spec)
    if not isinstance(round_num, int) or round_num < 1:
        return max_t
    min_t = MIN_TEMP_HIGH if max_t == MAX_TEMP_HIGH else MIN_TEMP_LOW
    step = (max_t - min_t) / (max_refinement_rounds - 1) \
        if max_refinement_rounds > 1 else 0.0
    return round(max(min_t, max_t - (round_num - 1) * step), 1)
