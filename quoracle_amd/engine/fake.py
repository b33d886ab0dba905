"""FakeEngine: deterministic scripted model backend for orchestrator tests.

The MI355X analogue of the reference's three-tier fake backends:
scripted token streams through the production path.  Supports:
  * a per-model queue of scripted responses,
  * a response function (model_key, messages, request) -> str,
  * deterministic bag-of-ngrams embeddings so semantic rules behave sensibly,
  * failure injection (errors, context overflow) per model.
"""

from __future__ import annotations

import asyncio
import hashlib
import json
import math
from collections import defaultdict, deque
from typing import Any, Callable, Dict, List, Optional, Sequence

from .api import Engine, GenerateRequest, GenerateResult

EMBED_DIM = 64


def deterministic_embedding(text: str, dim: int = EMBED_DIM) -> List[float]:
    """Hash-bucketed word-bigram embedding: similar texts get similar vectors,
    stable across runs."""
    vec = [0.0] * dim
    words = text.lower().split()
    for w in words:
        h = int(hashlib.md5(w.encode()).hexdigest()[:8], 16)
        vec[h % dim] += 1.0
    for a, b in zip(words, words[1:]):
        h = int(hashlib.md5(f"{a}|{b}".encode()).hexdigest()[:8], 16)
        vec[h % dim] += 0.5
    norm = math.sqrt(sum(v * v for v in vec)) or 1.0
    return [v / norm for v in vec]


def default_action_response(reasoning: str = "scripted", action: str = "wait",
                            params: Optional[dict] = None,
                            wait: Any = False) -> str:
    return json.dumps({
        "reasoning": reasoning,
        "action": action,
        "params": params or {},
        "wait": wait,
    })


class FakeEngine:
    """Engine implementation returning scripted responses."""

    def __init__(
        self,
        responses: Optional[Dict[str, List[str]]] = None,
        response_fn: Optional[Callable[[str, List[Dict[str, str]], GenerateRequest], str]] = None,
        default_response: Optional[str] = None,
        context_limits: Optional[Dict[str, int]] = None,
        output_limits: Optional[Dict[str, int]] = None,
        latency_s: float = 0.0,
    ):
        self._queues: Dict[str, deque] = defaultdict(deque)
        for model, texts in (responses or {}).items():
            self._queues[model].extend(texts)
        self._response_fn = response_fn
        self._default = default_response or default_action_response()
        self._context_limits = context_limits or {}
        self._output_limits = output_limits or {}
        self._latency_s = latency_s
        self._fail: Dict[str, str] = {}
        self.calls: List[GenerateRequest] = []
        self.embed_calls: List[List[str]] = []

    # -- failure injection ---------------------------------------------------
    def fail_model(self, model_key: str, reason: str = "simulated_failure") -> None:
        self._fail[model_key] = reason

    def heal_model(self, model_key: str) -> None:
        self._fail.pop(model_key, None)

    def push_response(self, model_key: str, text: str) -> None:
        self._queues[model_key].append(text)

    # -- Engine protocol -----------------------------------------------------
    async def generate(self, request: GenerateRequest) -> GenerateResult:
        self.calls.append(request)
        if self._latency_s:
            await asyncio.sleep(self._latency_s)
        reason = self._fail.get(request.model_key)
        if reason:
            return GenerateResult(model_key=request.model_key, error=reason)
        input_tokens = sum(self.count_tokens(m.get("content", ""))
                           for m in request.messages)
        if input_tokens > self.context_limit(request.model_key):
            return GenerateResult(model_key=request.model_key,
                                  error="context_overflow",
                                  input_tokens=input_tokens)
        if self._queues.get(request.model_key):
            text = self._queues[request.model_key].popleft()
        elif self._response_fn is not None:
            text = self._response_fn(request.model_key, request.messages, request)
        else:
            text = self._default
        return GenerateResult(
            model_key=request.model_key,
            text=text,
            input_tokens=input_tokens,
            output_tokens=self.count_tokens(text),
        )

    async def embed(self, texts: List[str]) -> Sequence[Sequence[float]]:
        return self.embed_sync(texts)

    def embed_sync(self, texts: List[str]) -> Sequence[Sequence[float]]:
        """Synchronous embedding path: the consensus merge rules call this
        inline (on GPU it is a direct engine forward, no event loop needed)."""
        self.embed_calls.append(list(texts))
        return [deterministic_embedding(t) for t in texts]

    def similarity_matrix(self, texts: List[str]):
        vecs = [deterministic_embedding(t) for t in texts]
        self.embed_calls.append(list(texts))
        out = []
        for a in vecs:
            out.append([sum(x * y for x, y in zip(a, b)) for b in vecs])
        return out

    def count_tokens(self, text: str) -> int:
        return max(1, len(text) // 4) if text else 0

    def context_limit(self, model_key: str) -> int:
        return self._context_limits.get(model_key, 128_000)

    def output_limit(self, model_key: str) -> int:
        return self._output_limits.get(model_key, 16_384)


def sync_embed_many(engine: Any) -> Callable[[List[str]], Sequence[Sequence[float]]]:
    """Get the synchronous EmbedManyFn the merge rules expect.

    Engines hosting the (small) embedding model locally expose embed_sync;
    falling back to a private event loop only happens off the main loop.
    """
    if hasattr(engine, "embed_sync"):
        return engine.embed_sync

    def _embed(texts: List[str]):
        try:
            asyncio.get_running_loop()
        except RuntimeError:
            return asyncio.run(engine.embed(texts))
        import concurrent.futures
        with concurrent.futures.ThreadPoolExecutor(max_workers=1) as ex:
            return ex.submit(lambda: asyncio.run(engine.embed(texts))).result()
    return _embed
