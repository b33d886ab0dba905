"""Paged-KV block management + prefix-cached sessions.

The reference scales context logically (per-model histories + condensation,
reference: agent/token_manager.ex, per_model_query/condensation.ex); here the
physical analogue is a paged KV cache sized for 288 GB HBM3E per GPU:

  * BlockManager — free-list allocator over fixed-size KV blocks shared by
    all sequences of one hosted model.
  * Session — a persistent conversation whose KV survives between generate
    calls: on the next call the engine diffs the new prompt against the
    cached token prefix and only prefills the tail (the MI355X replacement
    for the reference's provider prompt cache, consensus_handler.ex:126-152).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional


class OutOfBlocks(Exception):
    pass


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int):
        self.block_size = block_size
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def free_blocks(self) -> int:
        return len(self._free)

    def alloc(self, n: int) -> List[int]:
        if n > len(self._free):
            raise OutOfBlocks(f"need {n} blocks, {len(self._free)} free")
        return [self._free.pop() for _ in range(n)]

    def free(self, blocks: List[int]) -> None:
        self._free.extend(blocks)

    def blocks_for_tokens(self, n_tokens: int) -> int:
        return (n_tokens + self.block_size - 1) // self.block_size


@dataclass
class Session:
    """One persistent conversation's KV state for one hosted model."""
    session_id: str
    token_ids: List[int] = field(default_factory=list)   # tokens whose KV is cached
    blocks: List[int] = field(default_factory=list)

    def slot(self, pos: int, block_size: int) -> int:
        return self.blocks[pos // block_size] * block_size + pos % block_size


class SessionCache:
    """session_id -> Session with LRU eviction under block pressure."""

    def __init__(self, mgr: BlockManager):
        self.mgr = mgr
        self._sessions: Dict[str, Session] = {}
        self._lru: List[str] = []

    def get_or_create(self, session_id: str) -> Session:
        sess = self._sessions.get(session_id)
        if sess is None:
            sess = Session(session_id)
            self._sessions[session_id] = sess
        self._touch(session_id)
        return sess

    def _touch(self, session_id: str) -> None:
        if session_id in self._lru:
            self._lru.remove(session_id)
        self._lru.append(session_id)

    def drop(self, session_id: str) -> None:
        sess = self._sessions.pop(session_id, None)
        if sess:
            self.mgr.free(sess.blocks)
            sess.blocks = []
            sess.token_ids = []
        if session_id in self._lru:
            self._lru.remove(session_id)

    def match_prefix(self, sess: Session, prompt: List[int]) -> int:
        """Longest common prefix of the cached tokens and the new prompt;
        trailing cached tokens (a diverged tail) are discarded."""
        n = 0
        limit = min(len(sess.token_ids), len(prompt))
        while n < limit and sess.token_ids[n] == prompt[n]:
            n += 1
        if n < len(sess.token_ids):
            keep_blocks = self.mgr.blocks_for_tokens(n)
            self.mgr.free(sess.blocks[keep_blocks:])
            sess.blocks = sess.blocks[:keep_blocks]
            sess.token_ids = sess.token_ids[:n]
        return n

    def extend(self, sess: Session, new_tokens: List[int],
               active: Optional[List[str]] = None) -> List[int]:
        """Allocate blocks for new tokens (evicting idle LRU sessions under
        pressure) and record them; returns the global slots."""
        bs = self.mgr.block_size
        start = len(sess.token_ids)
        need_blocks = self.mgr.blocks_for_tokens(start + len(new_tokens)) \
            - len(sess.blocks)
        while need_blocks > self.mgr.free_blocks:
            victim = next((sid for sid in self._lru
                           if sid != sess.session_id
                           and (active is None or sid not in active)), None)
            if victim is None:
                raise OutOfBlocks(
                    f"KV pool exhausted: need {need_blocks} blocks, "
                    f"{self.mgr.free_blocks} free, no evictable session")
            self.drop(victim)
        if need_blocks > 0:
            sess.blocks.extend(self.mgr.alloc(need_blocks))
        sess.token_ids.extend(new_tokens)
        return [sess.slot(start + i, bs) for i in range(len(new_tokens))]
