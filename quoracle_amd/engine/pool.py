"""EnginePool: logical model keys -> engines.

A pool maps each logical model in a profile's model_pool (e.g.
"llama3-8b#0", "llama3-8b#1") to the Engine that hosts it — the local GPU
engine, a remote rank's engine via the control plane, or a FakeEngine in
tests.  The embedder (used by consensus merge rules and lesson dedup) is a
single designated engine, normally the local one so the cosine vote kernel
runs on the orchestrator's GPU.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import hashlib
import time as _time

from .api import Engine
from .fake import sync_embed_many

# reference parity: embeddings.ex:24-25,402-426 — SHA-keyed cache,
# 1h TTL, 1000 entries, LRU eviction of the oldest 10% under pressure
EMBED_CACHE_TTL_S = 3600.0
EMBED_CACHE_MAX = 1000


class EmbedCache:
    def __init__(self, ttl_s: float = EMBED_CACHE_TTL_S,
                 max_entries: int = EMBED_CACHE_MAX):
        self.ttl_s = ttl_s
        self.max_entries = max_entries
        self._data = {}          # sha -> (ts, vector)
        self.hits = 0
        self.misses = 0

    @staticmethod
    def key(text: str) -> str:
        return hashlib.sha256(text.encode("utf-8", "replace")).hexdigest()

    def get(self, text: str):
        k = self.key(text)
        entry = self._data.get(k)
        if entry is None or _time.monotonic() - entry[0] > self.ttl_s:
            self.misses += 1
            self._data.pop(k, None)
            return None
        self.hits += 1
        # LRU touch
        self._data[k] = (_time.monotonic(), entry[1])
        return entry[1]

    def put(self, text: str, vector) -> None:
        if len(self._data) >= self.max_entries:
            evict = max(1, self.max_entries // 10)
            for k in sorted(self._data, key=lambda k: self._data[k][0])[:evict]:
                self._data.pop(k, None)
        self._data[self.key(text)] = (_time.monotonic(), vector)


class EnginePool:
    def __init__(
        self,
        default: Optional[Engine] = None,
        by_model: Optional[Dict[str, Engine]] = None,
        embedder: Optional[Engine] = None,
    ):
        self._default = default
        self._by_model = by_model or {}
        self._embedder = embedder or default or next(iter(self._by_model.values()), None)
        self._embed_cache = EmbedCache()

    def engine_for(self, model_key: str) -> Engine:
        engine = self._by_model.get(model_key, self._default)
        if engine is None:
            raise KeyError(f"no engine hosts model {model_key}")
        return engine

    def assign(self, model_key: str, engine: Engine) -> None:
        self._by_model[model_key] = engine

    @property
    def embedder(self) -> Engine:
        if self._embedder is None:
            raise RuntimeError("no embedding engine configured")
        return self._embedder

    def embed_many_sync(self, texts: List[str]) -> Sequence[Sequence[float]]:
        return self.embed_facade(texts)

    @property
    def embed_facade(self) -> "EmbedFacade":
        return EmbedFacade(self.embedder, self._embed_cache)

    def models(self) -> List[str]:
        return list(self._by_model)

    def engine_stats(self) -> Dict[str, int]:
        """Aggregate telemetry over every distinct engine in the pool
        (LocalEngine.stats counters summed; fakes/remotes without stats
        contribute nothing). Feeds /metrics and /api/engine/stats."""
        engines = []
        for e in [self._default, self._embedder, *self._by_model.values()]:
            if e is not None and e not in engines:
                engines.append(e)
        total: Dict[str, int] = {}
        for e in engines:
            for key, val in (getattr(e, "stats", None) or {}).items():
                if isinstance(val, (int, float)):
                    total[key] = total.get(key, 0) + val
        return total


class EmbedFacade:
    """Callable embed_many with an optional fused similarity_matrix —
    components that only batch-compare texts (lesson dedup, clustering)
    use ONE embedding forward + cosine kernel pass instead of embedding
    then pairwise python cosine."""

    def __init__(self, engine, cache: Optional[EmbedCache] = None):
        self._engine = engine
        self._embed = sync_embed_many(engine)
        self._cache = cache
        # embedding-cost accumulator (reference: §2.6 — merged-rule
        # embedding calls accumulate, the agent flushes once per cycle);
        # cache hits are free, only fresh embeddings count
        self.embedded_tokens = 0

    def _count(self, text: str) -> int:
        """Exact token count via the engine's own tokenizer (the byte
        tokenizer is one call away — never estimate with len//4)."""
        try:
            return max(1, self._engine.count_tokens(text))
        except Exception:  # noqa: BLE001 — engines without a tokenizer
            return max(1, len(text) // 4)

    def __call__(self, texts: List[str]):
        texts = list(texts)
        if self._cache is None:
            return self._embed(texts)
        out = [None] * len(texts)
        missing = []
        for i, t in enumerate(texts):
            vec = self._cache.get(t)
            if vec is None:
                missing.append(i)
            else:
                out[i] = vec
        if missing:
            fresh = self._embed([texts[i] for i in missing])
            for i, vec in zip(missing, fresh):
                out[i] = vec
                self._cache.put(texts[i], vec)
            self.embedded_tokens += sum(
                self._count(texts[i]) for i in missing)
        return out

    @property
    def has_similarity(self) -> bool:
        return hasattr(self._engine, "similarity_matrix")

    def similarity_matrix(self, texts: List[str]):
        self.embedded_tokens += sum(self._count(t) for t in texts)
        if self.has_similarity:
            return self._engine.similarity_matrix(list(texts))
        from ..consensus.rules import cosine_similarity
        vecs = self(list(texts))
        return [[cosine_similarity(a, b) for b in vecs] for a in vecs]
