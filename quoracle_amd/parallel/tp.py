"""Tensor parallelism: head-sharded attention + row/col-parallel FFN.

BASELINE config 5 hosts Llama-3-70B and Mixtral-8x7B as TP=4 shards: each
rank of a TP group holds n_heads/tp query heads, n_kv_heads/tp KV heads
(with its own paged-KV shard), a column slice of wqkv/gate_up and a row
slice of wo/down.  Two RCCL all-reduces per layer (after the attention
projection and after the FFN down projection) restore the replicated
hidden state — the xGMI-native pattern, not a copy of
any NCCL call structure.

Determinism makes TP cheap to orchestrate: embeddings, norms, residuals,
logits and sampling are replicated, so every rank of a group runs the SAME
engine loop in lockstep and samples identical tokens from identical seeds —
no per-token broadcast.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class TPContext:
    rank: int
    world: int
    group: Optional[object] = None     # torch.distributed ProcessGroup

    def all_reduce_(self, x: torch.Tensor) -> torch.Tensor:
        """In-place sum-reduce across the TP group.  gloo (CPU tests) lacks
        bf16 — round-trip through fp32 there; RCCL reduces bf16 natively."""
        if self.world <= 1:
            return x
        if x.is_cuda or x.dtype not in (torch.bfloat16, torch.float16):
            dist.all_reduce(x, group=self.group)
            return x
        f = x.float()
        dist.all_reduce(f, group=self.group)
        x.copy_(f.to(x.dtype))
        return x


def shard_cols(w: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    """Column (output-dim) slice for col-parallel layers."""
    n = w.shape[1]
    assert n % world == 0, f"cannot shard {n} cols over {world}"
    step = n // world
    return w[:, rank * step:(rank + 1) * step].contiguous()


def shard_rows(w: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    """Row (input-dim) slice for row-parallel layers."""
    n = w.shape[0]
    assert n % world == 0, f"cannot shard {n} rows over {world}"
    step = n // world
    return w[rank * step:(rank + 1) * step].contiguous()


def shard_qkv(wqkv: torch.Tensor, n_heads: int, n_kv: int, head_dim: int,
              rank: int, world: int) -> torch.Tensor:
    """Slice the fused [hidden, (Hq+2*Hkv)*D] QKV weight by heads."""
    hq, hkv = n_heads // world, n_kv // world
    q = wqkv[:, :n_heads * head_dim]
    k = wqkv[:, n_heads * head_dim:(n_heads + n_kv) * head_dim]
    v = wqkv[:, (n_heads + n_kv) * head_dim:]
    return torch.cat([
        q[:, rank * hq * head_dim:(rank + 1) * hq * head_dim],
        k[:, rank * hkv * head_dim:(rank + 1) * hkv * head_dim],
        v[:, rank * hkv * head_dim:(rank + 1) * hkv * head_dim],
    ], dim=1).contiguous()


def shard_gate_up(w: torch.Tensor, intermediate: int, rank: int,
                  world: int) -> torch.Tensor:
    """gate and up halves are sharded independently so the local swiglu
    stays [gate_local | up_local]."""
    step = intermediate // world
    gate = w[..., rank * step:(rank + 1) * step]
    up = w[..., intermediate + rank * step:intermediate + (rank + 1) * step]
    return torch.cat([gate, up], dim=-1).contiguous()
