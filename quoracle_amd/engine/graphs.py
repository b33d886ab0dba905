"""hipGraph capture of the steady-state decode step.

The decode hot loop is hundreds of tiny kernels (32 layers x ~8 launches for
a 1-token-per-sequence batch): eager launch costs ~15 ms of host time per
step — far more than the GPU work.  Capture the whole decode forward +
logits GEMM into a hipGraph per (model, batch-bucket) with static
input/output buffers; each step then copies the new token/position/slot/
block-table values into the static buffers and replays one graph
("hipGraph capture of the steady-state decode step").

Batch sizes are bucketed to powers of two; rows beyond the real batch are
padded with ctx_len=1 pointing at a reserved scratch block, so padding work
is negligible and harmless (its KV writes land in the scratch block).

Prefill steps stay eager: they are few, large, and shape-diverse.
"""

from __future__ import annotations

import sys
from typing import Dict, List, Optional

import torch

from ..models.llama import ForwardBatch

_BUCKETS = [1, 2, 4, 8, 16, 32, 64]
NS_GRAPH = 32          # fixed split count inside the graph


def bucket_for(n: int) -> Optional[int]:
    for b in _BUCKETS:
        if n <= b:
            return b
    return None


class DecodeGraphs:
    def __init__(self, model, kv, device: torch.device, max_blocks_per_seq: int,
                 scratch_block: int):
        self.model = model
        self.kv = kv
        self.device = device
        self.maxb = max_blocks_per_seq
        self.scratch_block = scratch_block
        self.graphs: Dict[int, dict] = {}
        self.enabled = device.type == "cuda" and not bool(
            __import__("os").environ.get("QUORACLE_NO_GRAPHS"))
        self.pool = None
        # capture only happens while the GPU is quiescent (engine start);
        # a missing bucket at run time falls back to eager, never captures
        self.allow_capture = False

    # -- capture -------------------------------------------------------------

    def _capture(self, bucket: int) -> Optional[dict]:
        dev = self.device
        V = self.model.cfg.vocab_size
        bufs = {
            "tokens": torch.zeros(bucket, dtype=torch.int32, device=dev),
            "positions": torch.zeros(bucket, dtype=torch.int32, device=dev),
            "slots": torch.full((bucket,),
                                self.scratch_block * self.kv.block_size,
                                dtype=torch.int32, device=dev),
            "block_tables": torch.full((bucket, self.maxb), self.scratch_block,
                                       dtype=torch.int32, device=dev),
            "ctx_lens": torch.ones(bucket, dtype=torch.int32, device=dev),
        }
        rows = torch.arange(bucket, dtype=torch.long, device=dev)
        scope = __import__("os").environ.get("QUORACLE_GRAPH_SCOPE", "full")

        def _forward():
            batch = ForwardBatch(
                tokens=bufs["tokens"], positions=bufs["positions"],
                slots=bufs["slots"], block_tables=bufs["block_tables"],
                n_decode=bucket, ctx_lens=bufs["ctx_lens"],
                max_ctx=1 << 30)     # force the split decode path (NS fixed)
            hidden = self.model.forward(batch, self.kv)
            if scope == "noheads":
                return hidden
            return self.model.compute_logits(hidden, rows)

        try:
            # warmup on a side stream (required before capture)
            s = torch.cuda.Stream(dev)
            s.wait_stream(torch.cuda.current_stream(dev))
            with torch.cuda.stream(s):
                for _ in range(2):
                    logits = _forward()
            torch.cuda.current_stream(dev).wait_stream(s)
            torch.cuda.synchronize(dev)

            graph = torch.cuda.CUDAGraph()
            # thread_local: the orchestrator thread (embedder, monitors) may
            # issue work on other streams while this capture is open
            if self.pool is None:
                with torch.cuda.graph(graph, capture_error_mode="thread_local"):
                    logits = _forward()
                self.pool = graph.pool()
            else:
                with torch.cuda.graph(graph, pool=self.pool,
                                      capture_error_mode="thread_local"):
                    logits = _forward()
        except Exception as exc:  # noqa: BLE001 — graphs are an optimization
            print(f"[graphs] capture failed for bucket {bucket}: {exc}; "
                  f"falling back to eager decode", file=sys.stderr, flush=True)
            self.enabled = False
            return None
        pin = dev.type == "cuda"
        host = {
            "tokens": torch.zeros(bucket, dtype=torch.int32, pin_memory=pin),
            "positions": torch.zeros(bucket, dtype=torch.int32, pin_memory=pin),
            "slots": torch.zeros(bucket, dtype=torch.int32, pin_memory=pin),
            "ctx_lens": torch.ones(bucket, dtype=torch.int32, pin_memory=pin),
            "block_tables": torch.full((bucket, self.maxb), self.scratch_block,
                                       dtype=torch.int32, pin_memory=pin),
        }
        entry = {"graph": graph, "bufs": bufs, "logits": logits,
                 "rows": rows, "scope": scope, "host": host,
                 "bt_keys": [None] * bucket}
        self.graphs[bucket] = entry
        return entry

    def precapture(self, buckets=(1, 2, 4, 8, 16)) -> None:
        """Capture the decode graphs for the given batch buckets while the
        device is quiescent (called from engine start, before any traffic)."""
        if not self.enabled:
            return
        self.allow_capture = True
        try:
            for b in buckets:
                if b not in self.graphs and self._capture(b) is None:
                    break
        finally:
            self.allow_capture = False

    # -- replay --------------------------------------------------------------

    def run(self, tokens: List[int], positions: List[int], slots: List[int],
            bt_rows: List[List[int]], ctx_lens: List[int],
            bt_keys: Optional[List] = None):
        """Returns logits[:B] or None if graphs are unavailable.

        bt_keys: optional per-row identity (session_id, n_blocks) — rows
        whose key matches the staged copy skip the block-table rewrite
        (block lists only change every block_size tokens).
        """
        if not self.enabled:
            return None
        B = len(tokens)
        bucket = bucket_for(B)
        if bucket is None:
            return None
        if max(len(r) for r in bt_rows) > self.maxb:
            return None
        entry = self.graphs.get(bucket)
        if entry is None:
            if not self.allow_capture:
                return None
            entry = self._capture(bucket)
            if entry is None:
                return None
        bufs, host = entry["bufs"], entry["host"]
        pad_slot = self.scratch_block * self.kv.block_size
        # pinned staging: fill rows 0..B-1 (pad rows persist from capture),
        # then one async H2D per buffer
        host["tokens"][:B] = torch.tensor(tokens, dtype=torch.int32)
        host["tokens"][B:] = 0
        host["positions"][:B] = torch.tensor(positions, dtype=torch.int32)
        host["positions"][B:] = 0
        host["slots"][:B] = torch.tensor(slots, dtype=torch.int32)
        host["slots"][B:] = pad_slot
        host["ctx_lens"][:B] = torch.tensor(ctx_lens, dtype=torch.int32)
        host["ctx_lens"][B:] = 1
        bt_host = host["block_tables"]
        keys = entry["bt_keys"]
        bt_dirty = False
        for i, r in enumerate(bt_rows):
            key = bt_keys[i] if bt_keys else None
            if key is None or keys[i] != key:
                row = torch.full((self.maxb,), self.scratch_block,
                                 dtype=torch.int32)
                row[:len(r)] = torch.tensor(r, dtype=torch.int32)
                bt_host[i] = row
                keys[i] = key
                bt_dirty = True
        for i in range(len(bt_rows), bucket):
            if keys[i] is not None:
                bt_host[i] = self.scratch_block
                keys[i] = None
                bt_dirty = True
        for name in ("tokens", "positions", "slots", "ctx_lens"):
            bufs[name].copy_(host[name], non_blocking=True)
        if bt_dirty or bt_keys is None:
            bufs["block_tables"].copy_(bt_host, non_blocking=True)
        entry["graph"].replay()
        if entry["scope"] == "noheads":
            return self.model.compute_logits(entry["logits"],
                                             entry["rows"])[:B]
        return entry["logits"][:B]
