"""EnginePool: logical model keys -> engines.

A pool maps each logical model in a profile's model_pool (e.g.
"llama3-8b#0", "llama3-8b#1") to the Engine that hosts it — the local GPU
engine, a remote rank's engine via the control plane, or a FakeEngine in
tests.  The embedder (used by consensus merge rules and lesson dedup) is a
single designated engine, normally the local one so the cosine vote kernel
runs on the orchestrator's GPU.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence

from .api import Engine
from .fake import sync_embed_many


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
        return EmbedFacade(self.embedder)

    def models(self) -> List[str]:
        return list(self._by_model)


class EmbedFacade:
    """Callable embed_many with an optional fused similarity_matrix —
    components that only batch-compare texts (lesson dedup, clustering)
    use ONE embedding forward + cosine kernel pass instead of embedding
    then pairwise python cosine."""

    def __init__(self, engine):
        self._engine = engine
        self._embed = sync_embed_many(engine)

    def __call__(self, texts: List[str]):
        return self._embed(list(texts))

    @property
    def has_similarity(self) -> bool:
        return hasattr(self._engine, "similarity_matrix")

    def similarity_matrix(self, texts: List[str]):
        if self.has_similarity:
            return self._engine.similarity_matrix(list(texts))
        from ..consensus.rules import cosine_similarity
        vecs = self(list(texts))
        return [[cosine_similarity(a, b) for b in vecs] for a in vecs]
