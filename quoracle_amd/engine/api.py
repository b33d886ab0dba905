"""Engine interface: the seam between the orchestrator and model execution.

The reference's entire model-access layer is HTTP calls to remote providers
(reference: lib/quoracle/models/model_query.ex).  Here the same seam is an
async protocol implemented by:
  * LocalEngine   — HIP/CDNA4 inference on this process's GPU (engine/engine.py)
  * RemoteEngine  — a model hosted by another rank, reached over the gloo
                    control plane (parallel/control.py)
  * FakeEngine    — scripted responses for orchestrator tests (engine/fake.py)

All orchestrator code (consensus pipeline, agents, condensation) talks only to
this protocol, mirroring the reference's injectable model_query_fn/embedding_fn
test strategy.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Protocol, Sequence

# Dynamic max_tokens policy (reference: per_model_query.ex:17-24):
# reserve >= MIN_OUTPUT_TOKENS for output; assume tokenizer safety margin.
MIN_OUTPUT_TOKENS = 4096
TOKEN_SAFETY_MARGIN = 1.12


@dataclass
class GenerateRequest:
    model_key: str
    messages: List[Dict[str, str]]  # [{"role": ..., "content": ...}]
    temperature: float = 1.0
    max_tokens: int = MIN_OUTPUT_TOKENS
    seed: Optional[int] = None
    top_p: float = 1.0
    # Constrained decoding: force output to be a valid action JSON while the
    # model still does full forwards (the sampler masks logits per template).
    action_grammar: bool = False
    allowed_actions: Optional[List[str]] = None
    # grammar context: fills parameter literals the model cannot invent
    # (e.g. the child profile name for spawn_child)
    grammar_context: Optional[Dict[str, Any]] = None
    request_id: str = ""
    # Prefix-cache key: generate calls sharing a session_id reuse the KV of
    # the longest common token prefix (one session per (agent, model) —
    # the MI355X analogue of the reference's prompt cache,
    # reference: agent/consensus_handler.ex:126-152).
    session_id: str = ""


@dataclass
class GenerateResult:
    model_key: str
    text: str = ""
    input_tokens: int = 0
    output_tokens: int = 0
    latency_ms: float = 0.0
    error: Optional[str] = None           # "context_overflow" | other
    cost: float = 0.0

    @property
    def ok(self) -> bool:
        return self.error is None


class Engine(Protocol):
    """Async model-execution protocol."""

    async def generate(self, request: GenerateRequest) -> GenerateResult:
        ...

    async def embed(self, texts: List[str]) -> Sequence[Sequence[float]]:
        """Batch embedding used by the consensus vote / lesson dedup."""
        ...

    def count_tokens(self, text: str) -> int:
        ...

    def context_limit(self, model_key: str) -> int:
        ...

    def output_limit(self, model_key: str) -> int:
        ...


def dynamic_max_tokens(engine: Engine, model_key: str, input_tokens: int) -> int:
    """context_limit − margin×input, floored at MIN_OUTPUT_TOKENS and capped at
    the model's output limit (reference: per_model_query.ex:136-145)."""
    budget = engine.context_limit(model_key) - int(input_tokens * TOKEN_SAFETY_MARGIN)
    return min(max(budget, MIN_OUTPUT_TOKENS), engine.output_limit(model_key))


def count_messages_tokens(engine: Engine, messages: List[Dict[str, str]]) -> int:
    return sum(engine.count_tokens(m.get("content", "")) for m in messages)
