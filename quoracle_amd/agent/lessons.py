"""Lesson accumulation with embedding-based dedup.

Behavior-parity with the reference (reference: lib/quoracle/agent/
lesson_manager.ex:14-15,50-147): new lessons merge into existing ones when
embedding cosine similarity >= 0.90 (the survivor's confidence increments);
the per-model list is pruned to 100, lowest-confidence-oldest first.

On GPU the pairwise similarities come from one fused cosine-matrix kernel
(quoracle_amd.ops.cosine_sim_matrix) over the batch of lesson embeddings
instead of a per-pair loop.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..consensus.rules import EmbedManyFn, cosine_similarity

DEDUP_THRESHOLD = 0.90
MAX_LESSONS = 100


def merge_lessons(
    existing: List[Dict[str, Any]],
    new: List[Dict[str, Any]],
    embed_many: Optional[EmbedManyFn] = None,
    *,
    threshold: float = DEDUP_THRESHOLD,
    max_lessons: int = MAX_LESSONS,
) -> List[Dict[str, Any]]:
    """Merge `new` lessons into `existing` (most recent first).  Without an
    embedder, dedup falls back to exact text match."""
    merged = [dict(l) for l in existing]
    if not new:
        return merged[:max_lessons]

    sim = None
    existing_vecs = new_vecs = None
    if embed_many is not None and merged:
        texts = [l.get("text", "") for l in merged] + \
                [l.get("text", "") for l in new]
        if getattr(embed_many, "has_similarity", False):
            # one embedding forward + fused cosine kernel for ALL pairs
            sim = embed_many.similarity_matrix(texts)
        else:
            vectors = embed_many(texts)
            existing_vecs = vectors[: len(merged)]
            new_vecs = vectors[len(merged):]

    n_existing = len(merged)
    # rows[j] = this merged entry's row in the similarity matrix (sim path);
    # front-inserts during the loop keep both structures aligned
    rows = list(range(n_existing))
    for i, lesson in enumerate(new):
        duplicate_idx = None
        for j, old in enumerate(merged):
            if sim is not None:
                if sim[n_existing + i][rows[j]] >= threshold:
                    duplicate_idx = j
                    break
            elif existing_vecs is not None:
                s_ij = cosine_similarity(new_vecs[i], existing_vecs[j])
                if s_ij >= threshold:
                    duplicate_idx = j
                    break
            elif old.get("text") == lesson.get("text"):
                duplicate_idx = j
                break
        if duplicate_idx is not None:
            merged[duplicate_idx]["confidence"] = \
                merged[duplicate_idx].get("confidence", 1) + 1
        else:
            merged.insert(0, dict(lesson))
            rows.insert(0, n_existing + i)
            if existing_vecs is not None:
                existing_vecs = [new_vecs[i]] + list(existing_vecs)

    if len(merged) > max_lessons:
        # prune lowest-confidence, oldest (list is newest-first)
        order = sorted(range(len(merged)),
                       key=lambda k: (merged[k].get("confidence", 1), -k))
        drop = set(order[: len(merged) - max_lessons])
        merged = [l for k, l in enumerate(merged) if k not in drop]
    return merged
