"""Distributed consensus vote: cross-rank embedding all-gather over xGMI.

The consensus engine needs embeddings of candidate-action texts for
fingerprint clustering, semantic-similarity merges and lesson dedup
(SURVEY.md §2.10 P8).  With the pool sharded one-model-per-GPU, the
embedding work is spread round-robin over all ranks (each hosts a replica
of the small embed model) and the vectors come back in ONE RCCL all-gather
over xGMI — small latency-sensitive payloads on point-to-point links, the
exact case SURVEY.md §5.8 calls out.  The gather runs on the default
device process group (backend "nccl" = RCCL on ROCm); the control plane
only carries the text assignments.

Each server's engine thread keeps continuous-batching generation while its
main thread computes embeddings and joins the collective — the vote
overlaps the next prefill by construction.
"""

from __future__ import annotations

from typing import List, Sequence

import torch
import torch.distributed as dist

EMBED_DIM_PAD = 256      # embed-small hidden; fixed gather width


def compute_local_embeddings(engine, texts: List[str],
                             dim: int = EMBED_DIM_PAD) -> torch.Tensor:
    """[n, dim] normalized embeddings on the engine's device (fp32)."""
    if not texts:
        return torch.zeros((0, dim), device=engine.device)
    vecs = engine.embed_sync(texts)            # normalized python lists
    t = torch.tensor(vecs, dtype=torch.float32, device=engine.device)
    if t.shape[1] < dim:
        t = torch.nn.functional.pad(t, (0, dim - t.shape[1]))
    return t[:, :dim]


def all_gather_vote(local: torch.Tensor, counts: List[int],
                    group=None) -> List[torch.Tensor]:
    """All-gather per-rank embedding blocks padded to max(counts) rows.

    Returns the per-rank [counts[r], dim] tensors on every rank; over GPUs
    this is one RCCL all-gather on xGMI.
    """
    world = len(counts)
    maxn = max(counts) if counts else 0
    if maxn == 0:
        return [local[:0] for _ in range(world)]
    dim = local.shape[1]
    padded = torch.zeros((maxn, dim), dtype=torch.float32,
                         device=local.device)
    padded[:local.shape[0]] = local
    out = [torch.empty_like(padded) for _ in range(world)]
    dist.all_gather(out, padded, group=group)
    return [out[r][:counts[r]] for r in range(world)]


def assign_round_robin(texts: Sequence[str], world: int) -> List[List[str]]:
    shards: List[List[str]] = [[] for _ in range(world)]
    for i, t in enumerate(texts):
        shards[i % world].append(t)
    return shards


class DistributedEmbedder:
    """embed_many-compatible callable that spreads embedding compute over
    every rank and merges via the RCCL all-gather.  Falls back to the
    local engine when world == 1."""

    def __init__(self, engine, client, world: int, group=None):
        self.engine = engine        # rank-0 local engine (hosts embed model)
        self.client = client        # ControlClient or None
        self.world = world
        self.group = group

    def __call__(self, texts: List[str]):
        return self.embed_many(list(texts))

    # sync_embed_many (engine/fake.py) duck-types on this name
    def embed_sync(self, texts: List[str]):
        return self.embed_many(list(texts))

    def embed_many(self, texts: List[str]):
        if self.world <= 1 or self.client is None or len(texts) < self.world:
            return self.engine.embed_sync(texts)
        shards = assign_round_robin(texts, self.world)
        counts = [len(s) for s in shards]
        # tell every server its shard; they compute + join the all-gather
        self.client.embed_gather(shards, counts)
        local = compute_local_embeddings(self.engine, shards[0])
        blocks = all_gather_vote(local, counts, group=self.group)
        # un-interleave back to input order
        out = [None] * len(texts)
        for r, block in enumerate(blocks):
            rows = block.cpu().tolist()
            for j, row in enumerate(rows):
                out[r + j * self.world] = row
        return out
