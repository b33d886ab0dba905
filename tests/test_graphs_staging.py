"""CPU tests for the hipGraph replay staging logic (engine/graphs.py):
bucket selection, pad-row handling, and the bt_keys block-table staging
cache — including invalidation when eviction reuses block ids (the key
includes hash(tuple(blocks)), ROADMAP.md correctness backlog)."""

import types

import pytest
import torch

from quoracle_amd.engine.graphs import DecodeGraphs, bucket_for


def test_bucket_for():
    assert bucket_for(1) == 1
    assert bucket_for(3) == 4
    assert bucket_for(16) == 16
    assert bucket_for(17) == 32
    assert bucket_for(65) is None


class _FakeGraph:
    def __init__(self):
        self.replays = 0

    def replay(self):
        self.replays += 1


def _staged_graphs(bucket=4, maxb=6, scratch=99, block_size=16):
    """DecodeGraphs with a hand-built CPU entry (no capture, no GPU)."""
    g = DecodeGraphs.__new__(DecodeGraphs)
    g.kv = types.SimpleNamespace(block_size=block_size)
    g.maxb = maxb
    g.scratch_block = scratch
    g.enabled = True
    g.allow_capture = False
    fake = _FakeGraph()
    entry = {
        "graph": fake,
        "bufs": {
            "tokens": torch.zeros(bucket, dtype=torch.int32),
            "positions": torch.zeros(bucket, dtype=torch.int32),
            "slots": torch.zeros(bucket, dtype=torch.int32),
            "ctx_lens": torch.ones(bucket, dtype=torch.int32),
            "block_tables": torch.full((bucket, maxb), scratch,
                                       dtype=torch.int32),
        },
        "host": {
            "tokens": torch.zeros(bucket, dtype=torch.int32),
            "positions": torch.zeros(bucket, dtype=torch.int32),
            "slots": torch.zeros(bucket, dtype=torch.int32),
            "ctx_lens": torch.ones(bucket, dtype=torch.int32),
            "block_tables": torch.full((bucket, maxb), scratch,
                                       dtype=torch.int32),
        },
        "logits": torch.arange(bucket, dtype=torch.float32).unsqueeze(1),
        "rows": torch.arange(bucket),
        "scope": "full",
        "bt_keys": [None] * bucket,
    }
    g.graphs = {bucket: entry}
    g.pool = None
    return g, entry, fake


def test_replay_pads_rows_to_scratch_block():
    g, entry, fake = _staged_graphs()
    out = g.run(tokens=[5, 6, 7], positions=[10, 20, 30],
                slots=[160, 320, 480],
                bt_rows=[[1, 2], [3], [4]], ctx_lens=[33, 17, 9],
                bt_keys=[("s1", 101), ("s2", 202), ("s3", 303)])
    assert fake.replays == 1 and out.shape[0] == 3
    bufs = entry["bufs"]
    assert bufs["tokens"][:3].tolist() == [5, 6, 7]
    # pad row: slot -> scratch block, ctx 1, block table all-scratch
    assert bufs["slots"][3].item() == 99 * 16
    assert bufs["ctx_lens"][3:].tolist() == [1]
    assert (bufs["block_tables"][3] == 99).all()
    # real rows: blocks then scratch padding
    assert bufs["block_tables"][0][:3].tolist() == [1, 2, 99]


def test_bt_staging_cache_skips_rewrite_and_invalidates_on_change():
    g, entry, fake = _staged_graphs(bucket=1)
    key_a = ("s1", hash((1, 2)))
    g.run([5], [10], [160], [[1, 2]], [33], bt_keys=[key_a])
    staged = entry["host"]["block_tables"][0].clone()
    # poison the staged host row; an identical key must NOT rewrite it
    entry["host"]["block_tables"][0][0] = 77
    g.run([5], [11], [161], [[1, 2]], [34], bt_keys=[key_a])
    assert entry["host"]["block_tables"][0][0].item() == 77
    # eviction reuses block ids -> same session, different block list ->
    # different hash -> row is rewritten
    key_b = ("s1", hash((1, 4)))
    g.run([5], [12], [162], [[1, 4]], [35], bt_keys=[key_b])
    row = entry["host"]["block_tables"][0]
    assert row[0].item() == 1 and row[1].item() == 4
    assert entry["bt_keys"][0] == key_b


def test_pad_row_keys_reset_when_batch_shrinks():
    g, entry, fake = _staged_graphs()
    g.run([1, 2, 3, 4], [1, 2, 3, 4], [16, 32, 48, 64],
          [[1], [2], [3], [4]], [5, 5, 5, 5],
          bt_keys=[("a", 1), ("b", 2), ("c", 3), ("d", 4)])
    assert entry["bt_keys"][3] == ("d", 4)
    g.run([1, 2, 3], [5, 6, 7], [16, 32, 48], [[1], [2], [3]], [6, 6, 6],
          bt_keys=[("a", 1), ("b", 2), ("c", 3)])
    # row 3 is a pad row now: staged back to scratch, key cleared
    assert entry["bt_keys"][3] is None
    assert (entry["host"]["block_tables"][3] == 99).all()


def test_run_refuses_oversize_and_disabled():
    g, entry, fake = _staged_graphs(bucket=1, maxb=2)
    assert g.run([1], [1], [16], [[1, 2, 3]], [40]) is None   # > maxb
    assert g.run(list(range(70)), list(range(70)), list(range(70)),
                 [[1]] * 70, [1] * 70) is None                # > max bucket
    g.enabled = False
    assert g.run([1], [1], [16], [[1]], [4]) is None
    assert fake.replays == 0
